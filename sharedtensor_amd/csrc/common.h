// Common types for the MI355X-native shared-tensor engine.
//
// This is a from-scratch re-engineering of the capability set of
// Hello1024/shared-tensor (/root/reference/src/sharedtensor.c): a distributed
// shared tensor with approximate, compressed, error-feedback delta gossip over
// a self-organizing binary tree.  The reference is a single-threaded CPU/Lua
// extension; this engine keeps the replica and all delta staging in HBM3E,
// runs the codec as CDNA4 HIP kernels, and moves intra-node packets over RCCL
// p2p (xGMI) with TCP retained as control plane / inter-node data plane.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstring>
#include <deque>
#include <mutex>
#include <shared_mutex>
#include <string>
#include <thread>
#include <vector>

namespace shamd {

enum class Codec : uint32_t { OneBit = 0, Fp8 = 1, Int4 = 2 };

// Padded element count: every tensor's element space is padded to a multiple
// of 64 so that (a) a 64-lane wavefront never straddles a tensor boundary in
// the quantize kernel (__ballot packs one uint64 word per wave) and (b) all
// per-tensor payload regions are 8-byte aligned.
inline int64_t pad64(int64_t n) { return (n + 63) & ~int64_t(63); }

// Payload bytes for one tensor of n elements (matches ops/oracle.py).
inline int64_t payload_bytes(Codec c, int64_t n) {
  int64_t pe = pad64(n);
  switch (c) {
    case Codec::OneBit: return pe / 8;
    case Codec::Fp8: return pe;
    case Codec::Int4: return pe / 2;
  }
  return 0;
}

struct Config {
  std::string host;          // rendezvous host (master binds here)
  int port = 0;              // rendezvous port
  int device = -1;           // -1 = CPU, else HIP device ordinal
  Codec codec = Codec::OneBit;
  bool snapshot_join = true; // 'Y' is followed by a full state snapshot
  bool use_rccl = true;      // allow RCCL/xGMI upgrade for same-host GPU links
  bool reconnect = false;    // try to rejoin if the up-link dies
  bool preserve_subtree = false;  // on up-link death keep children attached
                             // and rejoin with the whole subtree: the
                             // reconciliation computes a correction delta
                             // (S + R - V_old) that drains to the children
                             // as ordinary gossip.  Acyclic by construction:
                             // the detached subtree is unreachable from the
                             // root's redirect walk.  Requires snapshot_join.
  double keepalive_s = 1.0;  // idle ping interval (reference: 1s, :161-164)
  double bw_limit = 0.0;     // bytes/sec cap per link, 0 = unlimited
                             // (reference TODO, README.md:31)
  double min_round_interval_s = 0.0;  // pace each link's send loop: the
                             // reference free-runs ("fills all bandwidth"),
                             // but on GPU the codec kernels share HBM with
                             // training compute — pacing trades (already
                             // tiny) staleness for compute bandwidth
  int expected_children = 2; // how many child links to provision buffers for
  std::vector<int64_t> sizes;  // per-tensor element counts (table mode;
                               // size 1 = classic flat tensor)
  std::string explicit_parent;  // "ip:port" — skip the discovery walk and
                                // connect straight to this node ("" = walk)
  int listen_port = 0;  // >0: bind the listener to this fixed port (explicit
                        // topologies); 0: reference behaviour — the local
                        // port of the up-connection
  double join_timeout_s = 60.0;
  int rms_sample_stride = 1;  // >1: subsample the RMS reduction (scale is a
                              // heuristic; stride k cuts its HBM traffic k×)
  bool lagged_scale = false;  // GPU: fold the scale statistic into the
                              // quantize kernel (next round's scale from this
                              // round's post-quantize residual) — removes the
                              // reduce pass from the steady state
  bool delta_bf16 = false;    // GPU: store per-link residual deltas in bf16
                              // (halves their HBM footprint; debit quanta
                              // are bf16-exact, only the remainder rounds)
  bool use_graphs = false;    // GPU: capture the per-round kernel+copy
                              // sequences into hipGraphs.  Measured NET
                              // NEGATIVE on MI355X/ROCm 7.2 (replay floor
                              // ~10-16us vs ~3.5us per plain launch; rounds
                              // are 2-4 ops) — kept as an option only.
};

struct LinkStatsSnap {
  uint64_t rounds_sent, rounds_recv, bytes_sent, bytes_recv;
  float last_scale_sent, last_scale_recv;
  bool active, dead;
  std::string peer;
  bool rccl;
};

constexpr uint32_t MAGIC = 0x53544132;  // "STA2"
constexpr uint16_t PROTO_VERSION = 2;

// client hello flags
constexpr uint16_t HELLO_HAS_GPU = 1;
constexpr uint16_t HELLO_WANT_RCCL = 2;
// accept flags
constexpr uint16_t ACC_SNAPSHOT = 1;
constexpr uint16_t ACC_RCCL = 2;

#pragma pack(push, 1)
struct Hello {           // client -> server right after connect
  uint32_t magic;
  uint16_t version;
  uint16_t flags;
  uint64_t n;            // total (unpadded) element count
  uint32_t ntensors;
  uint32_t codec;
  uint64_t hostid;       // for same-host (xGMI) detection
  int32_t device;        // HIP device ordinal (-1 = CPU)
  uint32_t pad;
};
struct AcceptHello {     // server -> client after 'Y'
  uint16_t version;
  uint16_t flags;
  uint32_t codec;        // authoritative codec for this link
  uint64_t n;
  uint32_t ntensors;
  uint32_t reserved;
};
struct PacketHeader {    // every TCP data-plane message
  uint8_t type;          // 0 = DATA, 1 = PING, 2 = CLOSE
  uint8_t codec;
  uint16_t reserved;
  uint32_t ntensors;
  // DATA: followed by fp32 scales[ntensors] + payload
};
#pragma pack(pop)

constexpr uint8_t PKT_DATA = 0;
constexpr uint8_t PKT_PING = 1;
constexpr uint8_t PKT_CLOSE = 2;

}  // namespace shamd
