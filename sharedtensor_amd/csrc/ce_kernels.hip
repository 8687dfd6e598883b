// Fused bf16 cross-entropy for the training path.
//
// torch's route (autocast CE on (R=B*T, V=50257) bf16 logits) runs separate
// softmax forward + backward kernels with fp32 intermediates — ~6% of a
// GPT-2-small step.  Here:
//   fwd: block per row, independent max pass + exp-sum pass (fp32 math,
//        fast-math exp — outputs are bf16-rounded anyway); emits per-row
//        loss and saves (max, logsumexp) for backward
//   bwd: one pass writing bf16 dlogits = g * (softmax - onehot)
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <stdexcept>
#include <string>

#include "hip_api.h"

namespace shamd {

#define HIP_CHECK_CE(expr)                                                 \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
  } while (0)

static __device__ __forceinline__ float ce_bf16_to_f32(uint16_t u) {
  return __uint_as_float(static_cast<uint32_t>(u) << 16);
}

static __device__ __forceinline__ uint16_t ce_f32_to_bf16(float f) {
  uint32_t u = __float_as_uint(f);
  if ((u & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return static_cast<uint16_t>(u >> 16);
}

static __device__ __forceinline__ float ce_pair_lo(uint32_t p) {
  return __uint_as_float(p << 16);
}
static __device__ __forceinline__ float ce_pair_hi(uint32_t p) {
  return __uint_as_float(p & 0xFFFF0000u);
}
static __device__ __forceinline__ uint32_t ce_pair_pack(float lo, float hi) {
  return static_cast<uint32_t>(ce_f32_to_bf16(lo)) |
         (static_cast<uint32_t>(ce_f32_to_bf16(hi)) << 16);
}

constexpr int CE_BLOCK = 256;  // 4 waves (fwd: measured vs 512, see launcher)

// block-wide sum (BLK/64-wave block): wave shfl tree + LDS combine
template <int BLK>
static __device__ __forceinline__ float block_sum(float v, float* ldsw) {
  for (int w = 32; w > 0; w >>= 1) v += __shfl_down(v, w, 64);
  int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) ldsw[wave] = v;
  __syncthreads();
  float t = 0.f;
#pragma unroll
  for (int i = 0; i < BLK / 64; ++i) t += ldsw[i];
  __syncthreads();
  return t;
}

// Rows are only 2-byte aligned when V is odd (GPT-2: V = 50257), so each
// row is processed as [optional head elem | aligned uint32 pairs | optional
// tail elem] — 4-byte loads halve the load-instruction count that bounds
// these kernels.
template <int BLK>
__global__ void k_ce_fwd(const uint16_t* __restrict__ logits,
                         const int32_t* __restrict__ targets,
                         float* __restrict__ loss, float* __restrict__ row_m,
                         float* __restrict__ row_lse, int64_t V) {
  // two independent passes (max, then sum of exp) — an online single pass
  // has a loop-carried (m, s) dependency per thread and measured 5x slower;
  // the second pass re-reads the row from L2 (rows are ~100 KB)
  __shared__ float lds4[BLK / 64];
  const int64_t r = blockIdx.x;
  const uint16_t* xr = logits + r * V;
  const int head = static_cast<int>(reinterpret_cast<uintptr_t>(xr) & 3) ? 1 : 0;
  const int64_t npairs = (V - head) >> 1;
  const bool tail = ((V - head) & 1) != 0;
  const uint32_t* xp = reinterpret_cast<const uint32_t*>(xr + head);

  float m = -INFINITY;
  for (int64_t p = threadIdx.x; p < npairs; p += BLK) {
    uint32_t u = xp[p];
    m = fmaxf(m, fmaxf(ce_pair_lo(u), ce_pair_hi(u)));
  }
  if (threadIdx.x == 0 && head) m = fmaxf(m, ce_bf16_to_f32(xr[0]));
  if (threadIdx.x == 1 && tail) m = fmaxf(m, ce_bf16_to_f32(xr[V - 1]));
  for (int w = 32; w > 0; w >>= 1) m = fmaxf(m, __shfl_down(m, w, 64));
  int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wave] = m;
  __syncthreads();
  float mm = -INFINITY;
#pragma unroll
  for (int i = 0; i < BLK / 64; ++i) mm = fmaxf(mm, lds4[i]);
  m = mm;
  __syncthreads();

  float s = 0.f;
  for (int64_t p = threadIdx.x; p < npairs; p += BLK) {
    uint32_t u = xp[p];
    s += __expf(ce_pair_lo(u) - m) + __expf(ce_pair_hi(u) - m);
  }
  if (threadIdx.x == 0 && head) s += __expf(ce_bf16_to_f32(xr[0]) - m);
  if (threadIdx.x == 1 && tail) s += __expf(ce_bf16_to_f32(xr[V - 1]) - m);
  float tot = block_sum<BLK>(s, lds4);
  if (threadIdx.x == 0) {
    float lse = __logf(tot) + m;
    float xt = ce_bf16_to_f32(xr[targets[r]]);
    loss[r] = lse - xt;
    row_m[r] = m;
    row_lse[r] = lse;
  }
}

template <int BLK>
__global__ void k_ce_bwd(const uint16_t* __restrict__ logits,
                         const int32_t* __restrict__ targets,
                         const float* __restrict__ row_lse,
                         uint16_t* __restrict__ dlogits,
                         const float* __restrict__ gscale_dev, float inv_r,
                         int64_t V) {
  const int64_t r = blockIdx.x;
  const uint16_t* xr = logits + r * V;
  uint16_t* dr = dlogits + r * V;
  float gscale = *gscale_dev * inv_r;  // upstream grad read on device: no
                                       // host sync in the backward pass
  float lse = row_lse[r];
  int64_t tgt = targets[r];
  const int head = static_cast<int>(reinterpret_cast<uintptr_t>(xr) & 3) ? 1 : 0;
  const int64_t npairs = (V - head) >> 1;
  const bool tail = ((V - head) & 1) != 0;
  const uint32_t* xp = reinterpret_cast<const uint32_t*>(xr + head);
  uint32_t* dp = reinterpret_cast<uint32_t*>(dr + head);
  for (int64_t p = threadIdx.x; p < npairs; p += BLK) {
    uint32_t u = xp[p];
    int64_t v0 = head + 2 * p;
    float g0 = gscale * (__expf(ce_pair_lo(u) - lse) - (v0 == tgt ? 1.f : 0.f));
    float g1 = gscale * (__expf(ce_pair_hi(u) - lse) - (v0 + 1 == tgt ? 1.f : 0.f));
    dp[p] = ce_pair_pack(g0, g1);
  }
  if (threadIdx.x == 0 && head) {
    float g = gscale * (__expf(ce_bf16_to_f32(xr[0]) - lse) - (0 == tgt ? 1.f : 0.f));
    dr[0] = ce_f32_to_bf16(g);
  }
  if (threadIdx.x == 1 && tail) {
    float g = gscale * (__expf(ce_bf16_to_f32(xr[V - 1]) - lse) -
                        (V - 1 == tgt ? 1.f : 0.f));
    dr[V - 1] = ce_f32_to_bf16(g);
  }
}

void hip_ce_fwd(const void* logits, const int32_t* targets, float* loss,
                float* row_m, float* row_lse, int64_t R, int64_t V,
                hipStream_t s) {
  // NOTE (measured, profiles/r02): an LDS-staged single-read variant
  // (whole 100 KB row resident in the 160 KiB LDS) ran 4x SLOWER than this
  // two-pass kernel -- at 100 KB/block only one block fits per CU and four
  // waves cannot hide the global-load latency; the second pass's L2 re-read
  // is cheaper than the lost occupancy.
  // block-size knob (SHTENS_CE_BLOCK=512); measured r02: 512 is within
  // noise of 256 (2.63 vs 2.66 ms fwd) — the kernel is bound by the
  // L2 re-read + exp throughput, not wave count — so 256 stays default
  const char* e = std::getenv("SHTENS_CE_BLOCK");
  if (e && std::atoi(e) == 512)
    hipLaunchKernelGGL(k_ce_fwd<512>, dim3(static_cast<uint32_t>(R)),
                       dim3(512), 0, s,
                       static_cast<const uint16_t*>(logits), targets, loss,
                       row_m, row_lse, V);
  else
    hipLaunchKernelGGL(k_ce_fwd<CE_BLOCK>, dim3(static_cast<uint32_t>(R)),
                       dim3(CE_BLOCK), 0, s,
                       static_cast<const uint16_t*>(logits), targets, loss,
                       row_m, row_lse, V);
  HIP_CHECK_CE(hipGetLastError());
}

void hip_ce_bwd(const void* logits, const int32_t* targets,
                const float* row_lse, void* dlogits, const float* gscale_dev,
                float inv_r, int64_t R, int64_t V, hipStream_t s) {
  const char* e = std::getenv("SHTENS_CE_BLOCK");
  if (e && std::atoi(e) == 512)
    hipLaunchKernelGGL(k_ce_bwd<512>, dim3(static_cast<uint32_t>(R)),
                       dim3(512), 0, s, static_cast<const uint16_t*>(logits),
                       targets, row_lse, static_cast<uint16_t*>(dlogits),
                       gscale_dev, inv_r, V);
  else
    hipLaunchKernelGGL(k_ce_bwd<CE_BLOCK>, dim3(static_cast<uint32_t>(R)),
                       dim3(CE_BLOCK), 0, s,
                       static_cast<const uint16_t*>(logits), targets,
                       row_lse, static_cast<uint16_t*>(dlogits), gscale_dev,
                       inv_r, V);
  HIP_CHECK_CE(hipGetLastError());
}

}  // namespace shamd
