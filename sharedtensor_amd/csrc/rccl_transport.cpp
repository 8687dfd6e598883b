#include "rccl_transport.h"

#include <rccl/rccl.h>

#include <chrono>
#include <cstring>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace shamd {

static_assert(sizeof(ncclUniqueId) == RCCL_ID_BYTES, "ncclUniqueId size");

#define NCCL_TRY(expr)                                                   \
  do {                                                                   \
    ncclResult_t _r = (expr);                                            \
    if (_r != ncclSuccess && _r != ncclInProgress)                       \
      throw std::runtime_error(std::string("RCCL error: ") +             \
                               ncclGetErrorString(_r));                  \
  } while (0)

struct RcclLink {
  ncclComm_t out = nullptr;  // communicator for this side's sends
  ncclComm_t in = nullptr;   // communicator for this side's recvs
  int peer = 0;              // peer rank within each 2-rank comm
  std::atomic<bool> aborted{false};
};

void rccl_make_ids(uint8_t ids[2 * RCCL_ID_BYTES]) {
  ncclUniqueId a, b;
  NCCL_TRY(ncclGetUniqueId(&a));
  NCCL_TRY(ncclGetUniqueId(&b));
  std::memcpy(ids, &a, RCCL_ID_BYTES);
  std::memcpy(ids + RCCL_ID_BYTES, &b, RCCL_ID_BYTES);
}

static void wait_comm(ncclComm_t c, double timeout_s, const char* what) {
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration_cast<std::chrono::steady_clock::duration>(
                      std::chrono::duration<double>(timeout_s));
  for (;;) {
    ncclResult_t st;
    NCCL_TRY(ncclCommGetAsyncError(c, &st));
    if (st == ncclSuccess) return;
    if (st != ncclInProgress)
      throw std::runtime_error(std::string("RCCL ") + what + " failed: " +
                               ncclGetErrorString(st));
    if (std::chrono::steady_clock::now() > deadline) {
      ncclCommAbort(c);
      throw std::runtime_error(std::string("RCCL ") + what + " timed out");
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(2));
  }
}

RcclLink* rccl_link_create(int device, const uint8_t ids[2 * RCCL_ID_BYTES],
                           bool is_parent, double timeout_s) {
  if (hipSetDevice(device) != hipSuccess)
    throw std::runtime_error("hipSetDevice failed in rccl_link_create");
  ncclUniqueId ida, idb;
  std::memcpy(&ida, ids, RCCL_ID_BYTES);
  std::memcpy(&idb, ids + RCCL_ID_BYTES, RCCL_ID_BYTES);
  int my_rank = is_parent ? 0 : 1;

  auto* l = new RcclLink();
  l->peer = 1 - my_rank;
  ncclConfig_t cfg = NCCL_CONFIG_INITIALIZER;
  cfg.blocking = 0;  // poll with timeout so a dead peer cannot hang us
  try {
    // direction A (ids[0]): parent -> child payloads; parent sends on it.
    // direction B (ids[1]): child -> parent payloads.
    ncclComm_t ca = nullptr, cb = nullptr;
    NCCL_TRY(ncclCommInitRankConfig(&ca, 2, ida, my_rank, &cfg));
    NCCL_TRY(ncclCommInitRankConfig(&cb, 2, idb, my_rank, &cfg));
    wait_comm(ca, timeout_s, "comm A init");
    wait_comm(cb, timeout_s, "comm B init");
    l->out = is_parent ? ca : cb;
    l->in = is_parent ? cb : ca;
  } catch (...) {
    delete l;
    throw;
  }
  return l;
}

// Our communicators are non-blocking (blocking=0), so ncclSend/ncclRecv may
// return ncclInProgress BEFORE the operation is enqueued on the stream —
// polling the stream alone would report "done" on an empty stream and let
// the caller reuse the buffer.  Order matters: (1) poll the comm's async
// state until the op is enqueued, (2) then poll the stream until it drains.
static bool wait_enqueued(RcclLink* l, ncclComm_t c,
                          const std::atomic<bool>& abort) {
  for (;;) {
    ncclResult_t st;
    if (ncclCommGetAsyncError(c, &st) != ncclSuccess) return false;
    if (st == ncclSuccess) return true;
    if (st != ncclInProgress) return false;
    if (abort.load() || l->aborted.load()) {
      ncclCommAbort(c);
      l->aborted.store(true);
      return false;
    }
    std::this_thread::sleep_for(std::chrono::microseconds(20));
  }
}

static bool wait_stream(RcclLink* l, ncclComm_t c, hipStream_t s,
                        const std::atomic<bool>& abort) {
  for (;;) {
    hipError_t e = hipStreamQuery(s);
    if (e == hipSuccess) return true;
    if (e != hipErrorNotReady) return false;
    ncclResult_t st;
    if (ncclCommGetAsyncError(c, &st) != ncclSuccess || (st != ncclSuccess && st != ncclInProgress))
      return false;
    if (abort.load() || l->aborted.load()) {
      ncclCommAbort(c);
      l->aborted.store(true);
      (void)hipStreamSynchronize(s);
      return false;
    }
    std::this_thread::sleep_for(std::chrono::microseconds(20));
  }
}

bool rccl_send(RcclLink* l, const void* buf, size_t bytes, hipStream_t stream,
               const std::atomic<bool>& abort) {
  if (l->aborted.load()) return false;
  ncclResult_t r = ncclSend(buf, bytes, ncclChar, l->peer, l->out, stream);
  if (r != ncclSuccess && r != ncclInProgress) return false;
  if (!wait_enqueued(l, l->out, abort)) return false;
  return wait_stream(l, l->out, stream, abort);
}

bool rccl_recv(RcclLink* l, void* buf, size_t bytes, hipStream_t stream,
               const std::atomic<bool>& abort) {
  if (l->aborted.load()) return false;
  ncclResult_t r = ncclRecv(buf, bytes, ncclChar, l->peer, l->in, stream);
  if (r != ncclSuccess && r != ncclInProgress) return false;
  if (!wait_enqueued(l, l->in, abort)) return false;
  return wait_stream(l, l->in, stream, abort);
}

void rccl_abort(RcclLink* l) {
  if (!l || l->aborted.exchange(true)) return;
  if (l->out) ncclCommAbort(l->out);
  if (l->in) ncclCommAbort(l->in);
}

void rccl_self_test(int device) {
  if (hipSetDevice(device) != hipSuccess)
    throw std::runtime_error("hipSetDevice failed");
  ncclUniqueId id;
  NCCL_TRY(ncclGetUniqueId(&id));
  ncclComm_t c = nullptr;
  ncclConfig_t cfg = NCCL_CONFIG_INITIALIZER;
  cfg.blocking = 0;
  NCCL_TRY(ncclCommInitRankConfig(&c, 1, id, 0, &cfg));
  wait_comm(c, 30.0, "self-test init");
  float* buf = nullptr;
  if (hipMalloc(&buf, 1024 * 4) != hipSuccess)
    throw std::runtime_error("hipMalloc failed");
  (void)hipMemset(buf, 0, 1024 * 4);
  hipStream_t s;
  (void)hipStreamCreate(&s);
  ncclResult_t r = ncclAllReduce(buf, buf, 1024, ncclFloat, ncclSum, c, s);
  if (r != ncclSuccess && r != ncclInProgress) {
    (void)hipFree(buf);
    throw std::runtime_error(std::string("self-test allreduce: ") +
                             ncclGetErrorString(r));
  }
  wait_comm(c, 30.0, "self-test allreduce enqueue");
  if (hipStreamSynchronize(s) != hipSuccess) {
    (void)hipFree(buf);
    throw std::runtime_error("self-test stream sync failed");
  }
  (void)hipStreamDestroy(s);
  (void)hipFree(buf);
  (void)ncclCommDestroy(c);
}

// Move a real payload through ncclSend/ncclRecv on one device: a 1-rank
// comm self-send/recv (grouped so the pair matches inside one kernel
// launch).  Exercises the non-blocking enqueue ordering (wait on comm async
// state, then the stream) with actual data movement and verifies the bytes.
// RCCL cannot make a 2-rank comm on ONE device ("Duplicate GPU detected",
// and aborting a half-made duplicate comm can hang), so this is the deepest
// payload-path execution a single leased GPU allows; the 2-GPU xGMI links
// use the identical rccl_send/rccl_recv code.
void rccl_loopback_payload(int device, int64_t bytes) {
  if (hipSetDevice(device) != hipSuccess)
    throw std::runtime_error("hipSetDevice failed");
  ncclUniqueId id;
  NCCL_TRY(ncclGetUniqueId(&id));
  ncclComm_t c = nullptr;
  ncclConfig_t cfg = NCCL_CONFIG_INITIALIZER;
  cfg.blocking = 0;
  NCCL_TRY(ncclCommInitRankConfig(&c, 1, id, 0, &cfg));
  wait_comm(c, 30.0, "loopback init");
  uint8_t *src = nullptr, *dst = nullptr;
  std::vector<uint8_t> host(static_cast<size_t>(bytes));
  for (int64_t i = 0; i < bytes; ++i)
    host[static_cast<size_t>(i)] = static_cast<uint8_t>(i * 131 + 7);
  hipStream_t s = nullptr;
  auto cleanup = [&] {
    if (s) (void)hipStreamDestroy(s);
    if (src) (void)hipFree(src);
    if (dst) (void)hipFree(dst);
    if (c) (void)ncclCommDestroy(c);
  };
  try {
    if (hipMalloc(&src, bytes) != hipSuccess ||
        hipMalloc(&dst, bytes) != hipSuccess)
      throw std::runtime_error("hipMalloc failed");
    if (hipMemcpy(src, host.data(), bytes, hipMemcpyHostToDevice) != hipSuccess)
      throw std::runtime_error("H2D failed");
    (void)hipMemset(dst, 0, bytes);
    if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess)
      throw std::runtime_error("stream create failed");
    NCCL_TRY(ncclGroupStart());
    NCCL_TRY(ncclSend(src, static_cast<size_t>(bytes), ncclChar, 0, c, s));
    NCCL_TRY(ncclRecv(dst, static_cast<size_t>(bytes), ncclChar, 0, c, s));
    ncclResult_t ge = ncclGroupEnd();
    if (ge != ncclSuccess && ge != ncclInProgress)
      throw std::runtime_error(std::string("group end: ") +
                               ncclGetErrorString(ge));
    wait_comm(c, 30.0, "loopback enqueue");  // same ordering as rccl_send
    if (hipStreamSynchronize(s) != hipSuccess)
      throw std::runtime_error("stream sync failed");
    std::vector<uint8_t> back(static_cast<size_t>(bytes));
    if (hipMemcpy(back.data(), dst, bytes, hipMemcpyDeviceToHost) != hipSuccess)
      throw std::runtime_error("D2H failed");
    if (back != host)
      throw std::runtime_error("loopback payload mismatch");
  } catch (...) {
    cleanup();
    throw;
  }
  cleanup();
}

void rccl_destroy(RcclLink* l) {
  if (!l) return;
  rccl_abort(l);
  if (l->out) ncclCommDestroy(l->out);
  if (l->in) ncclCommDestroy(l->in);
  delete l;
}

}  // namespace shamd
