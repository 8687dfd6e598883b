"""Fused bf16 RMSNorm (hand-written CDNA4 kernels, csrc/ln_kernels.hip).

Round-1 weak #5: the Llama RMSNorm module upcast the whole (B, T, C)
activation to fp32 per call — the same cast-traffic tax FusedLayerNorm
removed for GPT-2 (profiles/README.md).  These kernels keep tensors bf16
end-to-end with fp32 statistics: fwd saves rstd, backward is one
single-reduction dx pass plus a register-accumulated dgamma pass.
"""
from __future__ import annotations

import os

import torch

from .. import _core

MAX_C = 4096  # per-thread register accumulators in the dgamma kernel


class _FusedRMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        x = x.contiguous()
        C = x.shape[-1]
        R = x.numel() // C
        y = torch.empty_like(x)
        rstd = torch.empty(R, dtype=torch.float32, device=x.device)
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.rms_fwd(x.data_ptr(), weight.data_ptr(), y.data_ptr(),
                      rstd.data_ptr(), R, C, float(eps), s)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        C = x.shape[-1]
        R = x.numel() // C
        dx = torch.empty_like(x)
        dgamma = torch.zeros(C, dtype=torch.float32, device=x.device)
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.rms_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                      rstd.data_ptr(), dx.data_ptr(), dgamma.data_ptr(),
                      R, C, s)
        return dx, dgamma.to(w.dtype), None


def fused_rms_norm(x: torch.Tensor, weight: torch.Tensor,
                   eps: float) -> torch.Tensor:
    return _FusedRMSNormFn.apply(x, weight, eps)


def can_use(x: torch.Tensor, weight: torch.Tensor) -> bool:
    if os.environ.get("SHTENS_NO_FUSED_RMS") == "1":  # A/B measurement knob
        return False
    return (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and x.shape[-1] == weight.numel() and x.shape[-1] <= MAX_C
            and x.shape[-1] % 2 == 0)
