// Fused bf16 GELU (tanh approximation) forward/backward.
//
// Elementwise and HBM-bound; uint4 (8 x bf16) loads/stores and fast-math
// transcendentals (outputs are bf16-rounded anyway).
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "hip_api.h"

namespace shamd {

#define HIP_CHECK_GL(expr)                                                 \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
  } while (0)

static __device__ __forceinline__ float gl_lo(uint32_t p) {
  return __uint_as_float(p << 16);
}
static __device__ __forceinline__ float gl_hi(uint32_t p) {
  return __uint_as_float(p & 0xFFFF0000u);
}
static __device__ __forceinline__ uint16_t gl_bf16(float f) {
  uint32_t u = __float_as_uint(f);
  if ((u & 0x7FFFFFFFu) > 0x7F800000u) return 0x7FC0;
  u += 0x7FFFu + ((u >> 16) & 1u);
  return static_cast<uint16_t>(u >> 16);
}
static __device__ __forceinline__ uint32_t gl_pack(float lo, float hi) {
  return static_cast<uint32_t>(gl_bf16(lo)) |
         (static_cast<uint32_t>(gl_bf16(hi)) << 16);
}

// gelu(x) = 0.5 x (1 + tanh(k (x + 0.044715 x^3))), k = sqrt(2/pi)
constexpr float GK = 0.7978845608028654f;
constexpr float GC = 0.044715f;

static __device__ __forceinline__ float gelu_f(float x) {
  float inner = GK * (x + GC * x * x * x);
  return 0.5f * x * (1.f + tanhf(inner));
}

static __device__ __forceinline__ float gelu_grad_f(float x) {
  float x2 = x * x;
  float inner = GK * x * (1.f + GC * x2);
  float t = tanhf(inner);
  float sech2 = 1.f - t * t;
  return 0.5f * (1.f + t) + 0.5f * x * sech2 * GK * (1.f + 3.f * GC * x2);
}

// uint4 (8 bf16) per iteration — the pair-only variant measured SLOWER
// than torch's 8-wide elementwise kernels in round 1; round 2's SwiGLU
// showed uint4 vectorization is what closes (and flips) the gap.
__global__ void k_gelu_fwd(const uint4* __restrict__ x,
                           uint4* __restrict__ y, int64_t n8) {
  int64_t g = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8; i += g) {
    uint4 u = x[i];
    uint4 o;
    const uint32_t* up = &u.x;
    uint32_t* op = &o.x;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      op[k] = gl_pack(gelu_f(gl_lo(up[k])), gelu_f(gl_hi(up[k])));
    y[i] = o;
  }
}

__global__ void k_gelu_bwd(const uint4* __restrict__ dy,
                           const uint4* __restrict__ x,
                           uint4* __restrict__ dx, int64_t n8) {
  int64_t g = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n8; i += g) {
    uint4 u = x[i], d = dy[i];
    uint4 o;
    const uint32_t* up = &u.x;
    const uint32_t* dp = &d.x;
    uint32_t* op = &o.x;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      op[k] = gl_pack(gl_lo(dp[k]) * gelu_grad_f(gl_lo(up[k])),
                      gl_hi(dp[k]) * gelu_grad_f(gl_hi(up[k])));
    dx[i] = o;
  }
}

static inline int gelu_grid(int64_t npairs) {
  int64_t g = (npairs + 255) / 256;
  return static_cast<int>(g < 16384 ? (g > 0 ? g : 1) : 16384);
}

void hip_gelu_fwd(const void* x, void* y, int64_t n, hipStream_t s) {
  // wrapper guarantees n % 8 == 0 and 16-byte alignment
  if (n % 8) throw std::runtime_error("gelu: n must be a multiple of 8");
  hipLaunchKernelGGL(k_gelu_fwd, dim3(gelu_grid(n / 8)), dim3(256), 0, s,
                     static_cast<const uint4*>(x),
                     static_cast<uint4*>(y), n / 8);
  HIP_CHECK_GL(hipGetLastError());
}

void hip_gelu_bwd(const void* dy, const void* x, void* dx, int64_t n,
                  hipStream_t s) {
  if (n % 8) throw std::runtime_error("gelu: n must be a multiple of 8");
  hipLaunchKernelGGL(k_gelu_bwd, dim3(gelu_grid(n / 8)), dim3(256), 0, s,
                     static_cast<const uint4*>(dy),
                     static_cast<const uint4*>(x),
                     static_cast<uint4*>(dx), n / 8);
  HIP_CHECK_GL(hipGetLastError());
}

}  // namespace shamd
