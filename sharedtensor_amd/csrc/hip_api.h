// Host-callable launchers for the CDNA4 codec kernels (hip_kernels.hip).
//
// Element-space contract (shared with codec_cpu.cpp / ops/oracle.py):
//   * values/delta buffers are the UNPADDED concatenation of the table's
//     tensors (offs[t] element starts, offs[T] = n).
//   * the packed payload lives in PADDED element space: tensor t occupies
//     padded elements [poffs[t], poffs[t+1]), poffs multiples of 64, so a
//     64-lane wavefront never straddles a tensor and __ballot packs one
//     uint64 word per wave (LSB-first layout identical to the reference's
//     byte stream, sharedtensor.c:166-177).
//   * scales: fp32[T] at the head of the message buffer; payload starts at
//     align8(4*T).
#pragma once

#include <hip/hip_runtime.h>

#include "common.h"

namespace shamd {

inline int64_t align8(int64_t x) { return (x + 7) & ~int64_t(7); }

struct DevTable {
  int64_t* offs = nullptr;    // device, T+1 unpadded element offsets
  int64_t* poffs = nullptr;   // device, T+1 padded element offsets
  int T = 0;
  int64_t n = 0;              // offs[T]
  int64_t pe = 0;             // poffs[T]
};

// Residual delta buffers are fp32 (delta_bf16=false) or bf16
// (delta_bf16=true, halves delta HBM for 100 GB-scale tensors; debit quanta
// are exactly representable in bf16).  `delta`/`d1..d3` pointers are typed
// accordingly (void* at this interface).

// reduce_buf: device scratch, >= 8*T bytes (double sumsq for 1bit,
// fp32 absmax for fp8/int4). scales_out: device fp32[T].
void hip_reduce_scales(Codec c, const void* delta, bool delta_bf16,
                       const DevTable& tb, void* reduce_buf, float* scales_out,
                       int sample_stride, hipStream_t s);

// delta is debited in place (atomic, lossless vs concurrent adds);
// payload receives the packed bytes for the whole padded space.
// stats_out (nullable, 8*T bytes): accumulate post-quantize residual
// statistics for lagged-scale mode (zero it before the call).
void hip_quantize(Codec c, void* delta, bool delta_bf16, const DevTable& tb,
                  const float* scales_dev, uint8_t* payload, hipStream_t s,
                  void* stats_out = nullptr);

// Finalize scales from an already-populated reduce/stats buffer (lagged
// mode: the buffer was filled by the previous round's quantize).
void hip_finalize_scales(Codec c, const DevTable& tb, const void* reduce_buf,
                         float* scales_out, int sample_stride, hipStream_t s);

// Decode payload and accumulate into the fp32 replica (values, nullable)
// plus up to two gossip-forward residual buffers (sharedtensor.c:106-127).
void hip_apply(Codec c, const uint8_t* payload, const DevTable& tb,
               const float* scales_dev, float* values, void* d1, void* d2,
               bool delta_bf16, hipStream_t s);

// {values, d1..d3} += alpha * src over the flat unpadded space
// (addFromInternal, sharedtensor.c:334-344; alpha=-1 implements the
// snapshot debit).
void hip_add_scatter(const float* src, int64_t n, float alpha, float* values,
                     void* d1, void* d2, void* d3, bool delta_bf16,
                     hipStream_t s);

// Fused join-snapshot capture: out := values (atomic 32-bit loads) and
// delta -= out in one pass; `out` is the authoritative sent-bytes buffer.
void hip_snapshot_capture(const float* values, void* delta, bool delta_bf16,
                          float* out, int64_t n, hipStream_t s);

// {values, d1, d2} += src_delta (delta-typed source; rejoin reconciliation).
void hip_add_delta_scatter(const void* src_delta, bool delta_bf16, int64_t n,
                           float* values, void* d1, void* d2, hipStream_t s);

// Fused SGD-momentum update feeding the shared tensor: m = mu*m + g;
// u = -lr*m; {values, link deltas} += u.  One pass over HBM instead of four.
void hip_fused_sgd(float* mom, const float* grad, float lr, float momentum,
                   int64_t n, float* values, void* d1, void* d2, void* d3,
                   bool delta_bf16, hipStream_t s);

// Mixed-precision variant: bf16 grads in, fp32 master (values) updated,
// bf16 shadow params refreshed (folding concurrent gossip via the
// atomicAdd return), link deltas staged — all in one pass.
void hip_fused_sgd_bf16(float* mom, const uint16_t* grad, uint16_t* shadow,
                        float lr, float momentum, int64_t n, float* values,
                        void* d1, void* d2, void* d3, bool delta_bf16,
                        hipStream_t s);

// Fused AdamW feeding the shared tensor (torch.optim.AdamW semantics:
// decoupled weight decay on the pre-update weight): m/v fp32 state, grads
// fp32 or bf16, optional bf16 shadow refresh, update applied to values and
// staged into the link deltas — one HBM pass.  inv_bc* = 1/(1-beta*^t).
void hip_fused_adamw(float* mom, float* vel, const void* grad, bool grad_bf16,
                     uint16_t* shadow, float lr, float beta1, float beta2,
                     float eps, float wd, float inv_bc1, float inv_bc2,
                     int64_t n, float* values, void* d1, void* d2, void* d3,
                     bool delta_bf16, hipStream_t s);

// Fused bf16 LayerNorm for the training path (ln_kernels.hip): fwd saves
// fp32 mean/rstd; bwd = dx pass + register-accumulated dgamma/dbeta pass.
void hip_ln_fwd(const void* x, const void* w, const void* b, void* y,
                float* mean, float* rstd, int64_t R, int C, float eps,
                hipStream_t s);
void hip_ln_bwd(const void* dy, const void* x, const void* w,
                const float* mean, const float* rstd, void* dx, float* dgamma,
                float* dbeta, int64_t R, int C, hipStream_t s);

// Column sum of (R, C) bf16 into fp32 out[C] (Linear backward bias grad);
// caller zeroes `out` first.
void hip_colsum_bf16(const void* x, float* out, int64_t R, int C,
                     hipStream_t s);

// Fused bf16 RMSNorm for the Llama training path (ln_kernels.hip): fwd
// saves fp32 rstd; bwd = single-reduction dx pass + register-accumulated
// dgamma pass.
void hip_rms_fwd(const void* x, const void* w, void* y, float* rstd,
                 int64_t R, int C, float eps, hipStream_t s);
void hip_rms_bwd(const void* dy, const void* x, const void* w,
                 const float* rstd, void* dx, float* dgamma, int64_t R, int C,
                 hipStream_t s);

// Fused bf16 cross-entropy (ce_kernels.hip): online-logsumexp fwd saving
// per-row (max, lse); bwd writes bf16 dlogits = g*(softmax - onehot).
void hip_ce_fwd(const void* logits, const int32_t* targets, float* loss,
                float* row_m, float* row_lse, int64_t R, int64_t V,
                hipStream_t s);
void hip_ce_bwd(const void* logits, const int32_t* targets,
                const float* row_lse, void* dlogits, const float* gscale_dev,
                float inv_r, int64_t R, int64_t V, hipStream_t s);

// Fused bf16 SwiGLU (swiglu_kernels.hip): y = silu(x1)*x3 one kernel each
// way; n must be a multiple of 8, buffers 16-byte aligned (uint4 loads).
void hip_swiglu_fwd(const void* x1, const void* x3, void* y, int64_t n,
                    hipStream_t s);
void hip_swiglu_bwd(const void* dy, const void* x1, const void* x3, void* dx1,
                    void* dx3, int64_t n, hipStream_t s);

// Fused bf16 GELU-tanh (gelu_kernels.hip); n must be even, buffers
// 4-byte aligned (pair loads).
void hip_gelu_fwd(const void* x, void* y, int64_t n, hipStream_t s);
void hip_gelu_bwd(const void* dy, const void* x, void* dx, int64_t n,
                  hipStream_t s);

}  // namespace shamd
