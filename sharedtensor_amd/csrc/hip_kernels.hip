// CDNA4 (gfx950) kernels for the shared-tensor delta codec.
//
// These replace the reference's single-threaded CPU loops
// (/root/reference/src/sharedtensor.c:106-111, 156-174, 334-344) with
// HBM-bandwidth-bound GPU passes:
//   * k_reduce_*   — per-tensor RMS / absmax via 64-lane __shfl reduction +
//                    one atomic per wavefront
//   * k_quant_1bit — fused sign-quantize + error-feedback debit + __ballot
//                    bit-pack (one uint64 per wave, LSB-first layout
//                    byte-identical to the reference wire format)
//   * k_apply_*    — fused decode + multi-destination atomic scatter
//                    (values + gossip-forward buffers in one pass)
//   * k_fused_sgd  — optimizer update fused with the 4-way delta scatter
//
// All element loops are grid-stride with 256-thread blocks (4 waves); the
// kernels are HBM-bound, so the win is minimizing passes over memory, not
// MFMA work.  Wavefront size is 64 (CDNA4): __ballot returns uint64_t and a
// wave maps exactly onto one packed word.
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "hip_api.h"

namespace shamd {

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                  \
  } while (0)

constexpr int BLOCK = 256;

static inline int grid_for(int64_t n) {
  int64_t g = (n + BLOCK - 1) / BLOCK;
  // >> 256 workgroups fills all 8 XCDs; cap and grid-stride beyond.
  return static_cast<int>(g < 32768 ? (g > 0 ? g : 1) : 32768);
}

// ---------------------------------------------------------------- helpers

// Binary search: largest t with poffs[t] <= j.  T is small (1..few hundred)
// and the array is L2/LDS-hot; for T==1 the caller's fast path skips this.
__device__ __forceinline__ int find_tensor(const int64_t* poffs, int T, int64_t j) {
  int lo = 0, hi = T - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (poffs[mid] <= j) lo = mid; else hi = mid - 1;
  }
  return lo;
}

__device__ __forceinline__ float d_e4m3_to_f32(uint8_t v) {
  uint32_t exp = (v >> 3) & 0xF;
  uint32_t man = v & 7;
  float out;
  if (exp == 0xF && man == 7) {
    out = __builtin_nanf("");
  } else if (exp == 0) {
    out = static_cast<float>(man) * 0x1p-9f;
  } else {
    uint32_t bits = ((exp - 7 + 127) << 23) | (man << 20);
    out = __uint_as_float(bits);
  }
  return (v & 0x80) ? -out : out;
}

__device__ __forceinline__ uint8_t d_f32_to_e4m3(float xf) {
  if (xf != xf) return 0x7F;
  float cl = xf > 448.f ? 448.f : (xf < -448.f ? -448.f : xf);
  uint32_t u = __float_as_uint(cl);
  uint8_t sign = (u >> 24) & 0x80;
  uint32_t abs = u & 0x7FFFFFFFu;
  if (abs == 0) return sign;
  int e = static_cast<int>(abs >> 23) - 127;
  uint32_t mant = (abs & 0x7FFFFFu) | 0x800000u;
  int shift = (e >= -6) ? 20 : (14 - e);
  if (shift > 24) return sign;
  uint32_t q = mant >> shift;
  uint32_t rem = mant & ((1u << shift) - 1);
  uint32_t half = 1u << (shift - 1);
  if (rem > half || (rem == half && (q & 1))) q++;
  if (e >= -6) {
    if (q == 16) { q = 8; e++; }
    return sign | static_cast<uint8_t>(((e + 7) << 3) | (q - 8));
  }
  if (q >= 8) return sign | 0x08;
  return sign | static_cast<uint8_t>(q);
}

__device__ __forceinline__ float d_pow2_floor(double x) {
  if (!(x > 0.0) || isinf(x) || isnan(x)) return 0.0f;
  int e;
  frexp(x, &e);
  return static_cast<float>(ldexp(1.0, e - 1));
}

__device__ __forceinline__ float d_pow2_ceil(double x) {
  if (!(x > 0.0) || isinf(x) || isnan(x)) return 0.0f;
  int e;
  double m = frexp(x, &e);
  return static_cast<float>(ldexp(1.0, m == 0.5 ? e - 1 : e));
}

// ------------------------------------------------------------- reductions

// Sum of squares per tensor (1-bit codec).  Each lane accumulates into a
// register across its grid-stride iterations (a wave never straddles tensors
// within an iteration: poffs are 64-aligned, so the flush condition is
// wave-uniform); on tensor change and at the end, one __shfl tree + a single
// double atomic per wave.  This keeps the atomic count at ~waves, not
// ~elements/64 — the naive per-iteration atomic version measured 21 GB/s on
// MI355X (serialized f64 atomics on one address).
__device__ __forceinline__ void wave_flush_sumsq(double acc, int t, double* sumsq) {
  for (int w = 32; w > 0; w >>= 1) acc += __shfl_down(acc, w, 64);
  if ((threadIdx.x & 63) == 0 && acc != 0.0) atomicAdd(&sumsq[t], acc);
}

__global__ void k_reduce_sumsq(const float* __restrict__ delta,
                               const int64_t* offs, const int64_t* poffs,
                               int T, int64_t pe, int stride, double* sumsq) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  double acc = 0.0;
  int cur_t = -1;
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < pe; j += gstride) {
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j);
    if (t != cur_t) {
      if (cur_t >= 0) wave_flush_sumsq(acc, cur_t, sumsq);
      acc = 0.0;
      cur_t = t;
    }
    int64_t L = j - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    if (L < sz && (stride == 1 || (L % stride) == 0)) {
      double f = static_cast<double>(delta[offs[t] + L]);
      acc += f * f;
    }
  }
  if (cur_t >= 0) wave_flush_sumsq(acc, cur_t, sumsq);
}

// absmax per tensor (fp8/int4 codecs); same register-accumulate structure,
// float atomicMax via uint compare (valid for non-negative floats).
__device__ __forceinline__ void wave_flush_max(float acc, int t, uint32_t* amax) {
  for (int w = 32; w > 0; w >>= 1) {
    float o = __shfl_down(acc, w, 64);
    acc = o > acc ? o : acc;
  }
  if ((threadIdx.x & 63) == 0 && acc > 0.0f)
    atomicMax(&amax[t], __float_as_uint(acc));
}

__global__ void k_reduce_absmax(const float* __restrict__ delta,
                                const int64_t* offs, const int64_t* poffs,
                                int T, int64_t pe, uint32_t* amax) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  float acc = 0.0f;
  int cur_t = -1;
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < pe; j += gstride) {
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j);
    if (t != cur_t) {
      if (cur_t >= 0) wave_flush_max(acc, cur_t, amax);
      acc = 0.0f;
      cur_t = t;
    }
    int64_t L = j - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    if (L < sz) {
      float v = fabsf(delta[offs[t] + L]);
      acc = v > acc ? v : acc;
    }
  }
  if (cur_t >= 0) wave_flush_max(acc, cur_t, amax);
}

__global__ void k_finalize_scales(Codec c, const void* reduce_buf,
                                  const int64_t* offs, int T, int stride,
                                  float* scales) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= T) return;
  if (c == Codec::OneBit) {
    int64_t sz = offs[t + 1] - offs[t];
    int64_t cnt = (sz + stride - 1) / stride;
    double ss = reinterpret_cast<const double*>(reduce_buf)[t];
    scales[t] = d_pow2_floor(sqrt(ss / static_cast<double>(cnt > 0 ? cnt : 1)));
  } else {
    float mx = __uint_as_float(reinterpret_cast<const uint32_t*>(reduce_buf)[t]);
    double den = (c == Codec::Fp8) ? 448.0 : 7.0;
    scales[t] = (mx > 0.0f && !isinf(mx) && !isnan(mx))
                    ? d_pow2_ceil(static_cast<double>(mx) / den)
                    : 0.0f;
  }
}

// --------------------------------------------------------------- quantize

// Fused sign-quantize + error feedback + __ballot bit-pack.  The debit is an
// atomicAdd of -sent so concurrent adds (training thread, gossip forwards)
// landing between the read and the update are preserved — the GPU version of
// the reference's benign-race contract made lossless.
// stats_out (nullable): accumulate the POST-quantize residual's sum of
// squares per tensor (lagged-scale mode: next round's scale comes from this
// round's quantize, so the separate reduce pass disappears from the steady
// state).  Same register-accumulate + wave-flush structure as k_reduce_*.
__global__ void k_quant_1bit(float* __restrict__ delta, const int64_t* offs,
                             const int64_t* poffs, int T, int64_t pe,
                             const float* __restrict__ scales,
                             uint64_t* __restrict__ words, double* stats_out) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  double acc = 0.0;
  int cur_t = -1;
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < pe; j += gstride) {
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j);
    if (stats_out && t != cur_t) {
      if (cur_t >= 0) wave_flush_sumsq(acc, cur_t, stats_out);
      acc = 0.0;
      cur_t = t;
    }
    int64_t L = j - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    float s = scales[t];
    bool neg = false;  // bit value: 1 means -scale was sent
    if (L < sz) {
      float* p = delta + offs[t] + L;
      float v = __hip_atomic_load(reinterpret_cast<float*>(p), __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
      if (s != 0.0f) {
        neg = !(v > 0.0f);
        atomicAdd(p, neg ? s : -s);
      }
      if (stats_out) {
        // fp32 subtraction first: the statistic must see the same rounded
        // residual that lands in memory
        float rf = (s == 0.0f) ? v : (v - (neg ? -s : s));
        acc += static_cast<double>(rf) * static_cast<double>(rf);
      }
    }
    uint64_t mask = __ballot(neg);
    if ((threadIdx.x & 63) == 0) words[j >> 6] = mask;
  }
  if (stats_out && cur_t >= 0) wave_flush_sumsq(acc, cur_t, stats_out);
}

__global__ void k_quant_fp8(float* __restrict__ delta, const int64_t* offs,
                            const int64_t* poffs, int T, int64_t pe,
                            const float* __restrict__ scales,
                            uint8_t* __restrict__ payload, uint32_t* stats_out) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  float amax = 0.0f;
  int cur_t = -1;
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < pe; j += gstride) {
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j);
    if (stats_out && t != cur_t) {
      if (cur_t >= 0) wave_flush_max(amax, cur_t, stats_out);
      amax = 0.0f;
      cur_t = t;
    }
    int64_t L = j - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    float s = scales[t];
    uint8_t q = 0;
    if (L < sz) {
      float* p = delta + offs[t] + L;
      float v = __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      float sent = 0.0f;
      if (s != 0.0f) {
        q = d_f32_to_e4m3(v / s);
        sent = d_e4m3_to_f32(q) * s;
        atomicAdd(p, -sent);
      }
      if (stats_out) {
        float r = fabsf(v - sent);
        amax = r > amax ? r : amax;
      }
    }
    payload[j] = q;
  }
  if (stats_out && cur_t >= 0) wave_flush_max(amax, cur_t, stats_out);
}

__global__ void k_quant_int4(float* __restrict__ delta, const int64_t* offs,
                             const int64_t* poffs, int T, int64_t pe,
                             const float* __restrict__ scales,
                             uint8_t* __restrict__ payload, uint32_t* stats_out) {
  // one thread per payload byte = 2 elements (both in the same tensor:
  // padded regions are 64-aligned, hence even).  nb is only a multiple of
  // 32, so the loop runs to a 64-multiple with clamped indices to keep the
  // wave fully convergent for the stats __shfl flush.
  int64_t nb = pe / 2;
  int64_t nbp = (nb + 63) & ~int64_t(63);
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  float amax = 0.0f;
  int cur_t = -1;
  for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; b < nbp; b += gstride) {
    bool act = b < nb;
    int64_t j0 = (act ? b : nb - 1) * 2;
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j0);
    if (stats_out && t != cur_t) {
      if (cur_t >= 0) wave_flush_max(amax, cur_t, stats_out);
      amax = 0.0f;
      cur_t = t;
    }
    int64_t base = j0 - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    float s = scales[t];
    uint8_t byte = 0;
    for (int k = 0; k < 2; ++k) {
      int64_t L = base + k;
      if (act && L < sz) {
        float* p = delta + offs[t] + L;
        float v = __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        float sent = 0.0f;
        if (s != 0.0f) {
          float r = nearbyintf(v / s);
          r = r > 7.f ? 7.f : (r < -7.f ? -7.f : r);
          int8_t q = static_cast<int8_t>(r);
          byte |= static_cast<uint8_t>(q & 0xF) << (k * 4);
          sent = static_cast<float>(q) * s;
          atomicAdd(p, -sent);
        }
        if (stats_out) {
          float r2 = fabsf(v - sent);
          amax = r2 > amax ? r2 : amax;
        }
      }
    }
    if (act) payload[b] = byte;
  }
  if (stats_out && cur_t >= 0) wave_flush_max(amax, cur_t, stats_out);
}

// ------------------------------------------------------------------ apply

template <int CODEC>
__global__ void k_apply(const uint8_t* __restrict__ payload,
                        const int64_t* offs, const int64_t* poffs, int T,
                        int64_t pe, const float* __restrict__ scales,
                        float* d0, float* d1, float* d2, float* d3) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < pe; j += gstride) {
    int t = (T == 1) ? 0 : find_tensor(poffs, T, j);
    int64_t L = j - poffs[t];
    int64_t sz = offs[t + 1] - offs[t];
    if (L >= sz) continue;
    float s = scales[t];
    if (s == 0.0f) continue;
    float v;
    if (CODEC == 0) {
      uint64_t w = reinterpret_cast<const uint64_t*>(payload)[j >> 6];
      v = ((w >> (j & 63)) & 1) ? -s : s;
    } else if (CODEC == 1) {
      v = d_e4m3_to_f32(payload[j]) * s;
    } else {
      int8_t q = static_cast<int8_t>((payload[j >> 1] >> ((j & 1) * 4)) & 0xF);
      if (q > 7) q -= 16;
      v = static_cast<float>(q) * s;
      if (v == 0.0f) continue;
    }
    int64_t g = offs[t] + L;
    if (d0) atomicAdd(d0 + g, v);
    if (d1) atomicAdd(d1 + g, v);
    if (d2) atomicAdd(d2 + g, v);
    if (d3) atomicAdd(d3 + g, v);
  }
}

__global__ void k_add_scatter(const float* __restrict__ src, int64_t n,
                              float alpha, float* d0, float* d1, float* d2,
                              float* d3) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += gstride) {
    float v = alpha * src[i];
    if (v == 0.0f) continue;
    if (d0) atomicAdd(d0 + i, v);
    if (d1) atomicAdd(d1 + i, v);
    if (d2) atomicAdd(d2 + i, v);
    if (d3) atomicAdd(d3 + i, v);
  }
}

__device__ __forceinline__ float bf16_to_f32(uint16_t u) {
  return __uint_as_float(static_cast<uint32_t>(u) << 16);
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  uint32_t u = __float_as_uint(f);
  if ((u & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;  // NaN
  u += 0x7FFFu + ((u >> 16) & 1u);  // round-to-nearest-even
  return static_cast<uint16_t>(u >> 16);
}

// Mixed-precision fused optimizer: bf16 gradients in, fp32 master update,
// bf16 shadow parameters out — one HBM pass covering what autocast training
// otherwise spends three cast/copy kernel families on.  The atomicAdd's
// return value folds any concurrently-applied gossip into the shadow.
__global__ void k_fused_sgd_bf16(float* __restrict__ mom,
                                 const uint16_t* __restrict__ grad,
                                 uint16_t* __restrict__ shadow, float lr,
                                 float momentum, int64_t n,
                                 float* __restrict__ values, float* d1,
                                 float* d2, float* d3) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += gstride) {
    float m = momentum * mom[i] + bf16_to_f32(grad[i]);
    mom[i] = m;
    float u = -lr * m;
    float old = atomicAdd(values + i, u);
    shadow[i] = f32_to_bf16(old + u);
    if (d1) atomicAdd(d1 + i, u);
    if (d2) atomicAdd(d2 + i, u);
    if (d3) atomicAdd(d3 + i, u);
  }
}

__global__ void k_fused_sgd(float* __restrict__ mom,
                            const float* __restrict__ grad, float lr,
                            float momentum, int64_t n, float* d0, float* d1,
                            float* d2, float* d3) {
  int64_t gstride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n; i += gstride) {
    float m = momentum * mom[i] + grad[i];
    mom[i] = m;
    float u = -lr * m;
    if (d0) atomicAdd(d0 + i, u);
    if (d1) atomicAdd(d1 + i, u);
    if (d2) atomicAdd(d2 + i, u);
    if (d3) atomicAdd(d3 + i, u);
  }
}

// -------------------------------------------------------------- launchers

void hip_reduce_scales(Codec c, const float* delta, const DevTable& tb,
                       void* reduce_buf, float* scales_out, int stride,
                       hipStream_t s) {
  HIP_CHECK(hipMemsetAsync(reduce_buf, 0, tb.T * 8, s));
  // 2048 blocks = 8 per CU (full wave occupancy) while keeping the flush
  // atomic count at ~8K per launch; each wave register-accumulates across
  // its grid-stride iterations.
  int g = grid_for(tb.pe);
  if (g > 2048) g = 2048;
  if (c == Codec::OneBit) {
    hipLaunchKernelGGL(k_reduce_sumsq, dim3(g), dim3(BLOCK), 0, s, delta,
                       tb.offs, tb.poffs, tb.T, tb.pe, stride,
                       reinterpret_cast<double*>(reduce_buf));
  } else {
    hipLaunchKernelGGL(k_reduce_absmax, dim3(g), dim3(BLOCK), 0, s, delta,
                       tb.offs, tb.poffs, tb.T, tb.pe,
                       reinterpret_cast<uint32_t*>(reduce_buf));
  }
  int gt = (tb.T + BLOCK - 1) / BLOCK;
  hipLaunchKernelGGL(k_finalize_scales, dim3(gt), dim3(BLOCK), 0, s, c,
                     reduce_buf, tb.offs, tb.T, stride, scales_out);
  HIP_CHECK(hipGetLastError());
}

void hip_finalize_scales(Codec c, const DevTable& tb, const void* reduce_buf,
                         float* scales_out, int stride, hipStream_t s) {
  int gt = (tb.T + BLOCK - 1) / BLOCK;
  hipLaunchKernelGGL(k_finalize_scales, dim3(gt), dim3(BLOCK), 0, s, c,
                     reduce_buf, tb.offs, tb.T, stride, scales_out);
  HIP_CHECK(hipGetLastError());
}

void hip_quantize(Codec c, float* delta, const DevTable& tb,
                  const float* scales_dev, uint8_t* payload, hipStream_t s,
                  void* stats_out) {
  int g = grid_for(c == Codec::Int4 ? tb.pe / 2 : tb.pe);
  if (stats_out && g > 2048) g = 2048;  // bound the flush-atomic count
  switch (c) {
    case Codec::OneBit:
      hipLaunchKernelGGL(k_quant_1bit, dim3(g), dim3(BLOCK), 0, s, delta,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev,
                         reinterpret_cast<uint64_t*>(payload),
                         reinterpret_cast<double*>(stats_out));
      break;
    case Codec::Fp8:
      hipLaunchKernelGGL(k_quant_fp8, dim3(g), dim3(BLOCK), 0, s, delta,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev, payload,
                         reinterpret_cast<uint32_t*>(stats_out));
      break;
    case Codec::Int4:
      hipLaunchKernelGGL(k_quant_int4, dim3(g), dim3(BLOCK), 0, s, delta,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev, payload,
                         reinterpret_cast<uint32_t*>(stats_out));
      break;
  }
  HIP_CHECK(hipGetLastError());
}

void hip_apply(Codec c, const uint8_t* payload, const DevTable& tb,
               const float* scales_dev, float* d0, float* d1, float* d2,
               float* d3, hipStream_t s) {
  int g = grid_for(tb.pe);
  switch (c) {
    case Codec::OneBit:
      hipLaunchKernelGGL(k_apply<0>, dim3(g), dim3(BLOCK), 0, s, payload,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev, d0, d1, d2, d3);
      break;
    case Codec::Fp8:
      hipLaunchKernelGGL(k_apply<1>, dim3(g), dim3(BLOCK), 0, s, payload,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev, d0, d1, d2, d3);
      break;
    case Codec::Int4:
      hipLaunchKernelGGL(k_apply<2>, dim3(g), dim3(BLOCK), 0, s, payload,
                         tb.offs, tb.poffs, tb.T, tb.pe, scales_dev, d0, d1, d2, d3);
      break;
  }
  HIP_CHECK(hipGetLastError());
}

void hip_add_scatter(const float* src, int64_t n, float alpha, float* d0,
                     float* d1, float* d2, float* d3, hipStream_t s) {
  hipLaunchKernelGGL(k_add_scatter, dim3(grid_for(n)), dim3(BLOCK), 0, s, src,
                     n, alpha, d0, d1, d2, d3);
  HIP_CHECK(hipGetLastError());
}

void hip_fused_sgd(float* mom, const float* grad, float lr, float momentum,
                   int64_t n, float* d0, float* d1, float* d2, float* d3,
                   hipStream_t s) {
  hipLaunchKernelGGL(k_fused_sgd, dim3(grid_for(n)), dim3(BLOCK), 0, s, mom,
                     grad, lr, momentum, n, d0, d1, d2, d3);
  HIP_CHECK(hipGetLastError());
}

void hip_fused_sgd_bf16(float* mom, const uint16_t* grad, uint16_t* shadow,
                        float lr, float momentum, int64_t n, float* values,
                        float* d1, float* d2, float* d3, hipStream_t s) {
  hipLaunchKernelGGL(k_fused_sgd_bf16, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     mom, grad, shadow, lr, momentum, n, values, d1, d2, d3);
  HIP_CHECK(hipGetLastError());
}

}  // namespace shamd
