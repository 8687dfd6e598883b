// CPU implementation of the wire codecs (semantic twin of ops/oracle.py and
// of the HIP kernels in hip_kernels.hip).  Used for CPU-resident shared
// tensors (BASELINE config 1: plumbing on loopback without a GPU) and as the
// host-side reference in kernel numerics tests.
#pragma once

#include <cstdint>
#include <functional>

#include "common.h"

namespace shamd {

// Lock-free fp32 accumulate (the reference races plain float adds across
// threads, sharedtensor.c:334-344; we keep the async contract but make the
// races lossless with relaxed atomics).
inline void atomic_add_f32(float* p, float v) {
  uint32_t old = __atomic_load_n(reinterpret_cast<uint32_t*>(p), __ATOMIC_RELAXED);
  for (;;) {
    float f;
    __builtin_memcpy(&f, &old, 4);
    f += v;
    uint32_t nw;
    __builtin_memcpy(&nw, &f, 4);
    if (__atomic_compare_exchange_n(reinterpret_cast<uint32_t*>(p), &old, nw,
                                    true, __ATOMIC_RELAXED, __ATOMIC_RELAXED))
      return;
  }
}

inline float atomic_load_f32(const float* p) {
  uint32_t u = __atomic_load_n(reinterpret_cast<const uint32_t*>(p), __ATOMIC_RELAXED);
  float f;
  __builtin_memcpy(&f, &u, 4);
  return f;
}

// Chunked parallel-for over [0, n): fixed 4M-element chunks (64-aligned so
// packed-payload bytes are never shared across workers) executed by up to
// SHTENS_CPU_THREADS workers (default min(hw, 16)).  The chunk decomposition
// depends only on n, so chunk-local results are machine-independent; the
// reference's whole engine is one thread per link (sharedtensor.c) — this is
// where the CPU engine beats it by an order of magnitude on big tensors.
void cpu_pfor(int64_t n, const std::function<void(int64_t, int64_t)>& fn);

// OCP fp8 e4m3fn conversions (shared semantics with torch.float8_e4m3fn;
// round-to-nearest-even, caller pre-clamps to +-448).
uint8_t f32_to_e4m3(float x);
float e4m3_to_f32(uint8_t v);

// Power-of-two scale helpers (sharedtensor.c:159 semantics).
float pow2_floor_f(double x);
float pow2_ceil_f(double x);

// Per-tensor scale: 1bit -> 2^floor(log2(rms)); fp8 -> 2^ceil(log2(max/448));
// int4 -> 2^ceil(log2(max/7)).  `stride` subsamples the reduction (1 = exact).
float cpu_compute_scale(Codec c, const float* delta, int64_t n, int stride = 1);

// Fused quantize + pack + error feedback for one tensor region.
// `payload` receives payload_bytes(c, n) bytes; `delta` is debited in place
// with relaxed-atomic adds so concurrent accumulations are never lost.
void cpu_quantize(Codec c, float* delta, int64_t n, float scale, uint8_t* payload);

// Decode `payload` and accumulate the carried +-q*scale into every non-null
// destination (values + forwarded links, sharedtensor.c:106-127).
void cpu_apply(Codec c, const uint8_t* payload, int64_t n, float scale,
               float* const* dsts, int ndst);

// Plain fp32 scatter-add of src into every destination (addFromInternal,
// sharedtensor.c:334-344).
void cpu_add_scatter(const float* src, int64_t n, float* const* dsts, int ndst);

}  // namespace shamd
