// RCCL/xGMI data plane for same-host GPU link pairs.
//
// The reference moves every packet over TCP through host memory
// (sharedtensor.c:121-122,176-177).  On an MI355X node the 8 GPUs are fully
// connected by xGMI (7 p2p links x ~153 GB/s per GPU), so each tree edge
// whose endpoints are GPUs on the same host upgrades its data plane to a
// pair of 2-rank RCCL communicators (one per direction, so the send and
// recv threads never share a communicator).  The TCP connection stays as
// the control plane (join, keepalive, close, death detection).
//
// ncclUniqueIds are exchanged over the already-established TCP link, so the
// engine needs no external rendezvous (torch.distributed not required).
#pragma once

#include <hip/hip_runtime.h>

#include <atomic>
#include <cstdint>

namespace shamd {

constexpr int RCCL_ID_BYTES = 128;  // sizeof(ncclUniqueId)

struct RcclLink;  // opaque

// Parent generates two unique ids (one per direction) to ship over TCP.
void rccl_make_ids(uint8_t ids[2 * RCCL_ID_BYTES]);

// Both sides: create the two communicators (parent = rank 0 in both).
// Blocks until both sides join or timeout_s elapses (throws on failure).
RcclLink* rccl_link_create(int device, const uint8_t ids[2 * RCCL_ID_BYTES],
                           bool is_parent, double timeout_s);

// Post a send/recv of exactly `bytes` device bytes on `stream` and wait for
// completion, polling `abort` — returns false if aborted or errored.
bool rccl_send(RcclLink* l, const void* buf, size_t bytes, hipStream_t stream,
               const std::atomic<bool>& abort);
bool rccl_recv(RcclLink* l, void* buf, size_t bytes, hipStream_t stream,
               const std::atomic<bool>& abort);

void rccl_abort(RcclLink* l);    // break in-flight ops (idempotent)
void rccl_destroy(RcclLink* l);  // abort + free

// Single-GPU sanity check: size-1 non-blocking communicator + self
// all-reduce.  Validates librccl + the non-blocking poll path without a
// second GPU; throws on failure.
void rccl_self_test(int device);
void rccl_loopback_payload(int device, int64_t bytes);

}  // namespace shamd
