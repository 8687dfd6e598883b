// Fused bf16 SwiGLU for the Llama MLP: y = silu(x1) * x3 in ONE kernel.
//
// torch runs silu and mul as two elementwise kernels (5 tensor passes
// forward, and a 2-kernel backward); fusing them is a pure
// bandwidth win at (B*T, ffn_dim) activations:
//   fwd: read x1, x3; write y                        (3 passes vs 5)
//   bwd: read dy, x1, x3; write dx1, dx3             (5 passes vs 8)
// 8 bf16 elements per thread iteration via uint4 (16 B) loads — the same
// vectorization torch's elementwise kernels use (the pair-only fused GELU
// measured SLOWER than torch per element; see profiles/README.md round 1).
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "hip_api.h"

namespace shamd {

#define HIP_CHECK_SW(expr)                                                 \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
  } while (0)

static __device__ __forceinline__ float sw_lo(uint32_t p) {
  return __uint_as_float(p << 16);
}
static __device__ __forceinline__ float sw_hi(uint32_t p) {
  return __uint_as_float(p & 0xFFFF0000u);
}
static __device__ __forceinline__ uint16_t sw_bf16(float f) {
  uint32_t u = __float_as_uint(f);
  if ((u & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return static_cast<uint16_t>(u >> 16);
}
static __device__ __forceinline__ uint32_t sw_pack(float lo, float hi) {
  return static_cast<uint32_t>(sw_bf16(lo)) |
         (static_cast<uint32_t>(sw_bf16(hi)) << 16);
}
static __device__ __forceinline__ float sw_sigmoid(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

constexpr int SW_BLOCK = 256;

// n8 = n/8: each iteration handles one uint4 (8 bf16) from x1 and x3.
__global__ void k_swiglu_fwd(const uint4* __restrict__ x1,
                             const uint4* __restrict__ x3,
                             uint4* __restrict__ y, int64_t n8) {
  int64_t gs = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8;
       i += gs) {
    uint4 a = x1[i], b = x3[i];
    uint4 o;
    const uint32_t* ap = &a.x;
    const uint32_t* bp = &b.x;
    uint32_t* op = &o.x;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float a0 = sw_lo(ap[k]), a1 = sw_hi(ap[k]);
      float s0 = a0 * sw_sigmoid(a0), s1 = a1 * sw_sigmoid(a1);
      op[k] = sw_pack(s0 * sw_lo(bp[k]), s1 * sw_hi(bp[k]));
    }
    y[i] = o;
  }
}

// dx1 = dy * x3 * sig(x1) * (1 + x1 * (1 - sig(x1)));  dx3 = dy * silu(x1)
__global__ void k_swiglu_bwd(const uint4* __restrict__ dy,
                             const uint4* __restrict__ x1,
                             const uint4* __restrict__ x3,
                             uint4* __restrict__ dx1,
                             uint4* __restrict__ dx3, int64_t n8) {
  int64_t gs = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8;
       i += gs) {
    uint4 g = dy[i], a = x1[i], b = x3[i];
    uint4 o1, o3;
    const uint32_t* gp = &g.x;
    const uint32_t* ap = &a.x;
    const uint32_t* bp = &b.x;
    uint32_t* o1p = &o1.x;
    uint32_t* o3p = &o3.x;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float g0 = sw_lo(gp[k]), g1 = sw_hi(gp[k]);
      float a0 = sw_lo(ap[k]), a1 = sw_hi(ap[k]);
      float b0 = sw_lo(bp[k]), b1 = sw_hi(bp[k]);
      float sg0 = sw_sigmoid(a0), sg1 = sw_sigmoid(a1);
      o1p[k] = sw_pack(g0 * b0 * sg0 * (1.f + a0 * (1.f - sg0)),
                       g1 * b1 * sg1 * (1.f + a1 * (1.f - sg1)));
      o3p[k] = sw_pack(g0 * a0 * sg0, g1 * a1 * sg1);
    }
    dx1[i] = o1;
    dx3[i] = o3;
  }
}

static inline int sw_grid(int64_t n8) {
  int64_t g = (n8 + SW_BLOCK - 1) / SW_BLOCK;
  return static_cast<int>(g < 16384 ? (g > 0 ? g : 1) : 16384);
}

void hip_swiglu_fwd(const void* x1, const void* x3, void* y, int64_t n,
                    hipStream_t s) {
  if (n % 8) throw std::runtime_error("swiglu: n must be a multiple of 8");
  int64_t n8 = n / 8;
  hipLaunchKernelGGL(k_swiglu_fwd, dim3(sw_grid(n8)), dim3(SW_BLOCK), 0, s,
                     static_cast<const uint4*>(x1),
                     static_cast<const uint4*>(x3), static_cast<uint4*>(y),
                     n8);
  HIP_CHECK_SW(hipGetLastError());
}

void hip_swiglu_bwd(const void* dy, const void* x1, const void* x3, void* dx1,
                    void* dx3, int64_t n, hipStream_t s) {
  if (n % 8) throw std::runtime_error("swiglu: n must be a multiple of 8");
  int64_t n8 = n / 8;
  hipLaunchKernelGGL(k_swiglu_bwd, dim3(sw_grid(n8)), dim3(SW_BLOCK), 0, s,
                     static_cast<const uint4*>(dy),
                     static_cast<const uint4*>(x1),
                     static_cast<const uint4*>(x3), static_cast<uint4*>(dx1),
                     static_cast<uint4*>(dx3), n8);
  HIP_CHECK_SW(hipGetLastError());
}

}  // namespace shamd
