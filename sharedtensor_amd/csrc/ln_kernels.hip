// Fused bf16 LayerNorm forward/backward for the training path.
//
// torch's native LayerNorm backward on MI355X measured ~4.4x off the HBM
// roofline for GPT-2-small shapes (110 us per (32768, 768) call vs ~25 us
// of traffic; profiles/gpt2_train_step_kernels.txt).  These kernels do:
//   fwd:      one pass — block per row, fp32 statistics, bf16 in/out,
//             saves mean/rstd for backward
//   bwd dx:   one pass — block per row, two fused row-reductions
//   bwd dw/db: one pass — grid-stride over rows, per-thread column
//             accumulators in registers, one fp32 atomic per column per block
//
// Shapes: x is (R, C) row-major bf16, C <= 4096 (the wrapper falls back to
// torch otherwise), weights/bias bf16[C].
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "hip_api.h"  // declaration parity check

namespace shamd {

#define HIP_CHECK_LN(expr)                                                 \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
  } while (0)

static __device__ __forceinline__ float ln_bf16_to_f32(uint16_t u) {
  return __uint_as_float(static_cast<uint32_t>(u) << 16);
}

static __device__ __forceinline__ uint16_t ln_f32_to_bf16(float f) {
  uint32_t u = __float_as_uint(f);
  if ((u & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return static_cast<uint16_t>(u >> 16);
}

static __device__ __forceinline__ float bf16pair_lo(uint32_t p) {
  return __uint_as_float(p << 16);
}
static __device__ __forceinline__ float bf16pair_hi(uint32_t p) {
  return __uint_as_float(p & 0xFFFF0000u);
}
static __device__ __forceinline__ uint32_t bf16pair_pack(float lo, float hi) {
  return static_cast<uint32_t>(ln_f32_to_bf16(lo)) |
         (static_cast<uint32_t>(ln_f32_to_bf16(hi)) << 16);
}

constexpr int LN_BLOCK = 256;  // 4 waves

// Block-wide sum of one value per thread (4-wave block): wave shfl tree +
// LDS combine.  Returns the total to every thread.
static __device__ __forceinline__ float block_sum(float v, float* lds4) {
  for (int w = 32; w > 0; w >>= 1) v += __shfl_down(v, w, 64);
  int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wave] = v;
  __syncthreads();
  float t = lds4[0] + lds4[1] + lds4[2] + lds4[3];
  __syncthreads();
  return t;
}

// Per-thread register cache for a row's owned elements.  CPT (elements per
// thread) is a template parameter so the caches index statically and stay
// in registers — a runtime loop bound would push them to scratch memory.
constexpr int LN_MAXC = 4096;

template <int CPT>  // pairs per thread
__global__ void k_ln_fwd(const uint16_t* __restrict__ x,
                         const uint16_t* __restrict__ w,
                         const uint16_t* __restrict__ b,
                         uint16_t* __restrict__ y, float* __restrict__ mean,
                         float* __restrict__ rstd, int C, float eps) {
  __shared__ float lds4[4];
  const int64_t r = blockIdx.x;
  const int C2 = C >> 1;  // wrapper guarantees C is even
  const uint32_t* xr = reinterpret_cast<const uint32_t*>(x + r * C);
  const uint32_t* wp = reinterpret_cast<const uint32_t*>(w);
  const uint32_t* bp = reinterpret_cast<const uint32_t*>(b);
  uint32_t* yr = reinterpret_cast<uint32_t*>(y + r * C);
  float x0[CPT], x1[CPT];
  float s = 0.f, ss = 0.f;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    float a = 0.f, bb = 0.f;
    if (c < C2) {
      uint32_t pk = xr[c];
      a = bf16pair_lo(pk);
      bb = bf16pair_hi(pk);
    }
    x0[k] = a;
    x1[k] = bb;
    s += a + bb;
    ss += a * a + bb * bb;
  }
  float tot = block_sum(s, lds4);
  float tot2 = block_sum(ss, lds4);
  float m = tot / C;
  float var = tot2 / C - m * m;
  float rs = rsqrtf(var > 0.f ? var + eps : eps);
  if (threadIdx.x == 0) {
    mean[r] = m;
    rstd[r] = rs;
  }
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C2) {
      uint32_t wk = wp[c];
      float o0 = (x0[k] - m) * rs * bf16pair_lo(wk);
      float o1 = (x1[k] - m) * rs * bf16pair_hi(wk);
      if (bp) {
        uint32_t bk = bp[c];
        o0 += bf16pair_lo(bk);
        o1 += bf16pair_hi(bk);
      }
      yr[c] = bf16pair_pack(o0, o1);
    }
  }
}

template <int CPT>
__global__ void k_ln_bwd_dx(const uint16_t* __restrict__ dy,
                            const uint16_t* __restrict__ x,
                            const uint16_t* __restrict__ w,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd,
                            uint16_t* __restrict__ dx, int C) {
  __shared__ float lds4[4];
  const int64_t r = blockIdx.x;
  const int C2 = C >> 1;
  const uint32_t* xr = reinterpret_cast<const uint32_t*>(x + r * C);
  const uint32_t* dyr = reinterpret_cast<const uint32_t*>(dy + r * C);
  const uint32_t* wp = reinterpret_cast<const uint32_t*>(w);
  uint32_t* dxr = reinterpret_cast<uint32_t*>(dx + r * C);
  float m = mean[r], rs = rstd[r];
  float a0[CPT], a1[CPT], h0[CPT], h1[CPT];
  float s1 = 0.f, s2 = 0.f;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    float u0 = 0.f, u1 = 0.f, v0 = 0.f, v1 = 0.f;
    if (c < C2) {
      uint32_t dk = dyr[c], wk = wp[c], xk = xr[c];
      u0 = bf16pair_lo(dk) * bf16pair_lo(wk);
      u1 = bf16pair_hi(dk) * bf16pair_hi(wk);
      v0 = (bf16pair_lo(xk) - m) * rs;
      v1 = (bf16pair_hi(xk) - m) * rs;
    }
    a0[k] = u0;
    a1[k] = u1;
    h0[k] = v0;
    h1[k] = v1;
    s1 += u0 + u1;
    s2 += u0 * v0 + u1 * v1;
  }
  float t1 = block_sum(s1, lds4) / C;
  float t2 = block_sum(s2, lds4) / C;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C2)
      dxr[c] = bf16pair_pack((a0[k] - t1 - h0[k] * t2) * rs,
                             (a1[k] - t1 - h1[k] * t2) * rs);
  }
}

// dgamma/dbeta: each thread owns the columns {tid, tid+256, ...} and
// accumulates them in registers across its grid-stride rows; one fp32
// atomic per owned column per block at the end.  (Scalar ownership: the
// pair-vectorized variant measured 2x slower here — partial-wave idling in
// the odd half-iteration outweighs the wider loads.)
template <int CPT>
__global__ void k_ln_bwd_dwdb(const uint16_t* __restrict__ dy,
                              const uint16_t* __restrict__ x,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              float* __restrict__ dgamma,
                              float* __restrict__ dbeta, int64_t R, int C) {
  float accg[CPT], accb[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) accg[k] = accb[k] = 0.f;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const uint16_t* xr = x + r * C;
    const uint16_t* dyr = dy + r * C;
    float m = mean[r], rs = rstd[r];
#pragma unroll
    for (int k = 0; k < CPT; ++k) {
      int c = threadIdx.x + k * LN_BLOCK;
      if (c < C) {
        float g = ln_bf16_to_f32(dyr[c]);
        float xh = (ln_bf16_to_f32(xr[c]) - m) * rs;
        accg[k] += g * xh;
        accb[k] += g;
      }
    }
  }
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C) {
      atomicAdd(&dgamma[c], accg[k]);
      atomicAdd(&dbeta[c], accb[k]);
    }
  }
}

// ------------------------------------------------------------- RMSNorm
// Fused bf16 RMSNorm for the Llama training path (round-1 weak #5: the
// module-level fp32 upcast re-read the whole (B,T,C) activation per call —
// the same cast-traffic tax FusedLayerNorm removed for GPT-2).
//   fwd:    y = x * rsqrt(mean(x^2) + eps) * w, one pass, saves fp32 rstd
//   bwd dx: dx = rs * (dy*w - x * rs^2 * mean(dy*w*x))  — ONE row reduction
//   bwd dw: grid-stride rows, register column accumulators (as LN dw/db)

template <int CPT>  // pairs per thread
__global__ void k_rms_fwd(const uint16_t* __restrict__ x,
                          const uint16_t* __restrict__ w,
                          uint16_t* __restrict__ y, float* __restrict__ rstd,
                          int C, float eps) {
  __shared__ float lds4[4];
  const int64_t r = blockIdx.x;
  const int C2 = C >> 1;
  const uint32_t* xr = reinterpret_cast<const uint32_t*>(x + r * C);
  const uint32_t* wp = reinterpret_cast<const uint32_t*>(w);
  uint32_t* yr = reinterpret_cast<uint32_t*>(y + r * C);
  float x0[CPT], x1[CPT];
  float ss = 0.f;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    float a = 0.f, b = 0.f;
    if (c < C2) {
      uint32_t pk = xr[c];
      a = bf16pair_lo(pk);
      b = bf16pair_hi(pk);
    }
    x0[k] = a;
    x1[k] = b;
    ss += a * a + b * b;
  }
  float tot2 = block_sum(ss, lds4);
  float rs = rsqrtf(tot2 / C + eps);
  if (threadIdx.x == 0) rstd[r] = rs;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C2) {
      uint32_t wk = wp[c];
      yr[c] = bf16pair_pack(x0[k] * rs * bf16pair_lo(wk),
                            x1[k] * rs * bf16pair_hi(wk));
    }
  }
}

template <int CPT>
__global__ void k_rms_bwd_dx(const uint16_t* __restrict__ dy,
                             const uint16_t* __restrict__ x,
                             const uint16_t* __restrict__ w,
                             const float* __restrict__ rstd,
                             uint16_t* __restrict__ dx, int C) {
  __shared__ float lds4[4];
  const int64_t r = blockIdx.x;
  const int C2 = C >> 1;
  const uint32_t* xr = reinterpret_cast<const uint32_t*>(x + r * C);
  const uint32_t* dyr = reinterpret_cast<const uint32_t*>(dy + r * C);
  const uint32_t* wp = reinterpret_cast<const uint32_t*>(w);
  uint32_t* dxr = reinterpret_cast<uint32_t*>(dx + r * C);
  float rs = rstd[r];
  float g0[CPT], g1[CPT], v0[CPT], v1[CPT];
  float s = 0.f;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    float u0 = 0.f, u1 = 0.f, a = 0.f, b = 0.f;
    if (c < C2) {
      uint32_t dk = dyr[c], wk = wp[c], xk = xr[c];
      u0 = bf16pair_lo(dk) * bf16pair_lo(wk);
      u1 = bf16pair_hi(dk) * bf16pair_hi(wk);
      a = bf16pair_lo(xk);
      b = bf16pair_hi(xk);
    }
    g0[k] = u0;
    g1[k] = u1;
    v0[k] = a;
    v1[k] = b;
    s += u0 * a + u1 * b;
  }
  float t = block_sum(s, lds4) / C * rs * rs;
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C2)
      dxr[c] = bf16pair_pack((g0[k] - v0[k] * t) * rs,
                             (g1[k] - v1[k] * t) * rs);
  }
}

template <int CPT>
__global__ void k_rms_bwd_dw(const uint16_t* __restrict__ dy,
                             const uint16_t* __restrict__ x,
                             const float* __restrict__ rstd,
                             float* __restrict__ dgamma, int64_t R, int C) {
  float accg[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) accg[k] = 0.f;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const uint16_t* xr = x + r * C;
    const uint16_t* dyr = dy + r * C;
    float rs = rstd[r];
#pragma unroll
    for (int k = 0; k < CPT; ++k) {
      int c = threadIdx.x + k * LN_BLOCK;
      if (c < C)
        accg[k] += ln_bf16_to_f32(dyr[c]) * ln_bf16_to_f32(xr[c]) * rs;
    }
  }
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C) atomicAdd(&dgamma[c], accg[k]);
  }
}

void hip_rms_fwd(const void* x, const void* w, void* y, float* rstd,
                 int64_t R, int C, float eps, hipStream_t s) {
  if (C > LN_MAXC || (C & 1)) throw std::runtime_error("rms: bad C");
  int cpt = (C / 2 + LN_BLOCK - 1) / LN_BLOCK;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(static_cast<uint32_t>(R)), dim3(LN_BLOCK), 0,
                       s, static_cast<const uint16_t*>(x),
                       static_cast<const uint16_t*>(w),
                       static_cast<uint16_t*>(y), rstd, C, eps);
  };
  if (cpt <= 1) launch(k_rms_fwd<1>);
  else if (cpt <= 2) launch(k_rms_fwd<2>);
  else if (cpt <= 3) launch(k_rms_fwd<3>);
  else if (cpt <= 4) launch(k_rms_fwd<4>);
  else if (cpt <= 8) launch(k_rms_fwd<8>);
  else launch(k_rms_fwd<16>);
  HIP_CHECK_LN(hipGetLastError());
}

void hip_rms_bwd(const void* dy, const void* x, const void* w,
                 const float* rstd, void* dx, float* dgamma, int64_t R, int C,
                 hipStream_t s) {
  if (C > LN_MAXC || (C & 1)) throw std::runtime_error("rms: bad C");
  int cpt = (C / 2 + LN_BLOCK - 1) / LN_BLOCK;
  auto launch_dx = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(static_cast<uint32_t>(R)), dim3(LN_BLOCK), 0,
                       s, static_cast<const uint16_t*>(dy),
                       static_cast<const uint16_t*>(x),
                       static_cast<const uint16_t*>(w), rstd,
                       static_cast<uint16_t*>(dx), C);
  };
  if (cpt <= 1) launch_dx(k_rms_bwd_dx<1>);
  else if (cpt <= 2) launch_dx(k_rms_bwd_dx<2>);
  else if (cpt <= 3) launch_dx(k_rms_bwd_dx<3>);
  else if (cpt <= 4) launch_dx(k_rms_bwd_dx<4>);
  else if (cpt <= 8) launch_dx(k_rms_bwd_dx<8>);
  else launch_dx(k_rms_bwd_dx<16>);
  HIP_CHECK_LN(hipGetLastError());
  int g = R < 2048 ? static_cast<int>(R) : 2048;  // 8 blocks/CU (see LN note)
  int ecpt = (C + LN_BLOCK - 1) / LN_BLOCK;
  auto launch_dw = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g), dim3(LN_BLOCK), 0, s,
                       static_cast<const uint16_t*>(dy),
                       static_cast<const uint16_t*>(x), rstd, dgamma, R, C);
  };
  if (ecpt <= 1) launch_dw(k_rms_bwd_dw<1>);
  else if (ecpt <= 2) launch_dw(k_rms_bwd_dw<2>);
  else if (ecpt <= 3) launch_dw(k_rms_bwd_dw<3>);
  else if (ecpt <= 4) launch_dw(k_rms_bwd_dw<4>);
  else if (ecpt <= 8) launch_dw(k_rms_bwd_dw<8>);
  else launch_dw(k_rms_bwd_dw<16>);
  HIP_CHECK_LN(hipGetLastError());
}

// Column sum of a (R, C) bf16 matrix into fp32 out[C] — the bias-gradient
// reduction of Linear backward (torch's reduce_kernel measured ~1.7 TB/s
// at GPT-2 shapes).  Same register-accumulator geometry as k_ln_bwd_dwdb:
// each thread owns columns {tid, tid+256, ...} across grid-stride rows,
// one fp32 atomic per owned column per block.
template <int CPT>
__global__ void k_colsum_bf16(const uint16_t* __restrict__ x,
                              float* __restrict__ out, int64_t R, int C) {
  float acc[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) acc[k] = 0.f;
  for (int64_t r = blockIdx.x; r < R; r += gridDim.x) {
    const uint16_t* xr = x + r * C;
#pragma unroll
    for (int k = 0; k < CPT; ++k) {
      int c = threadIdx.x + k * LN_BLOCK;
      if (c < C) acc[k] += ln_bf16_to_f32(xr[c]);
    }
  }
#pragma unroll
  for (int k = 0; k < CPT; ++k) {
    int c = threadIdx.x + k * LN_BLOCK;
    if (c < C) atomicAdd(&out[c], acc[k]);
  }
}

void hip_colsum_bf16(const void* x, float* out, int64_t R, int C,
                     hipStream_t s) {
  if (C > LN_MAXC) throw std::runtime_error("colsum: C too large");
  int g = R < 2048 ? static_cast<int>(R) : 2048;
  int ecpt = (C + LN_BLOCK - 1) / LN_BLOCK;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g), dim3(LN_BLOCK), 0, s,
                       static_cast<const uint16_t*>(x), out, R, C);
  };
  if (ecpt <= 1) launch(k_colsum_bf16<1>);
  else if (ecpt <= 2) launch(k_colsum_bf16<2>);
  else if (ecpt <= 3) launch(k_colsum_bf16<3>);
  else if (ecpt <= 4) launch(k_colsum_bf16<4>);
  else if (ecpt <= 8) launch(k_colsum_bf16<8>);
  else launch(k_colsum_bf16<16>);
  HIP_CHECK_LN(hipGetLastError());
}

void hip_ln_fwd(const void* x, const void* w, const void* b, void* y,
                float* mean, float* rstd, int64_t R, int C, float eps,
                hipStream_t s) {
  if (C > LN_MAXC || (C & 1)) throw std::runtime_error("ln: bad C");
  int cpt = (C / 2 + LN_BLOCK - 1) / LN_BLOCK;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(static_cast<uint32_t>(R)), dim3(LN_BLOCK), 0,
                       s, static_cast<const uint16_t*>(x),
                       static_cast<const uint16_t*>(w),
                       static_cast<const uint16_t*>(b),
                       static_cast<uint16_t*>(y), mean, rstd, C, eps);
  };
  if (cpt <= 1) launch(k_ln_fwd<1>);
  else if (cpt <= 2) launch(k_ln_fwd<2>);
  else if (cpt <= 3) launch(k_ln_fwd<3>);
  else if (cpt <= 4) launch(k_ln_fwd<4>);
  else if (cpt <= 8) launch(k_ln_fwd<8>);
  else launch(k_ln_fwd<16>);
  HIP_CHECK_LN(hipGetLastError());
}

void hip_ln_bwd(const void* dy, const void* x, const void* w,
                const float* mean, const float* rstd, void* dx, float* dgamma,
                float* dbeta, int64_t R, int C, hipStream_t s) {
  if (C > LN_MAXC || (C & 1)) throw std::runtime_error("ln: bad C");
  int cpt = (C / 2 + LN_BLOCK - 1) / LN_BLOCK;
  auto launch_dx = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(static_cast<uint32_t>(R)), dim3(LN_BLOCK), 0,
                       s, static_cast<const uint16_t*>(dy),
                       static_cast<const uint16_t*>(x),
                       static_cast<const uint16_t*>(w), mean, rstd,
                       static_cast<uint16_t*>(dx), C);
  };
  if (cpt <= 1) launch_dx(k_ln_bwd_dx<1>);
  else if (cpt <= 2) launch_dx(k_ln_bwd_dx<2>);
  else if (cpt <= 3) launch_dx(k_ln_bwd_dx<3>);
  else if (cpt <= 4) launch_dx(k_ln_bwd_dx<4>);
  else if (cpt <= 8) launch_dx(k_ln_bwd_dx<8>);
  else launch_dx(k_ln_bwd_dx<16>);
  HIP_CHECK_LN(hipGetLastError());
  // 2048 blocks = 8 per CU: enough waves to hide the strided bf16 loads
  // (256 blocks measured 0.67 TB/s — latency-bound at 1 block/CU); the
  // per-column atomic count stays trivial (2048 per column).
  int g = R < 2048 ? static_cast<int>(R) : 2048;
  int ecpt = (C + LN_BLOCK - 1) / LN_BLOCK;  // element (not pair) ownership
  auto launch_dw = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g), dim3(LN_BLOCK), 0, s,
                       static_cast<const uint16_t*>(dy),
                       static_cast<const uint16_t*>(x), mean, rstd, dgamma,
                       dbeta, R, C);
  };
  if (ecpt <= 1) launch_dw(k_ln_bwd_dwdb<1>);
  else if (ecpt <= 2) launch_dw(k_ln_bwd_dwdb<2>);
  else if (ecpt <= 3) launch_dw(k_ln_bwd_dwdb<3>);
  else if (ecpt <= 4) launch_dw(k_ln_bwd_dwdb<4>);
  else if (ecpt <= 8) launch_dw(k_ln_bwd_dwdb<8>);
  else launch_dw(k_ln_bwd_dwdb<16>);
  HIP_CHECK_LN(hipGetLastError());
}

}  // namespace shamd
