"""Fused bf16 LayerNorm (hand-written CDNA4 kernels, csrc/ln_kernels.hip).

torch's native LayerNorm backward measured ~4.4x off the HBM roofline on
MI355X for transformer shapes; this pair of kernels (one fwd pass, dx pass +
register-accumulated dgamma/dbeta pass) replaces it on the GPU training
path.  fp32 statistics, bf16 tensors in/out.
"""
from __future__ import annotations

import torch

from .. import _core

MAX_C = 4096  # per-thread register accumulators in the dw/db kernel


class _FusedLayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        C = x.shape[-1]
        R = x.numel() // C
        y = torch.empty_like(x)
        mean = torch.empty(R, dtype=torch.float32, device=x.device)
        rstd = torch.empty(R, dtype=torch.float32, device=x.device)
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.ln_fwd(x.data_ptr(), weight.data_ptr(),
                     bias.data_ptr() if bias is not None else 0,
                     y.data_ptr(), mean.data_ptr(), rstd.data_ptr(), R, C,
                     float(eps), s)
        ctx.save_for_backward(x, weight, mean, rstd)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        C = x.shape[-1]
        R = x.numel() // C
        dx = torch.empty_like(x)
        dwdb = torch.zeros(2 * C, dtype=torch.float32, device=x.device)
        dgamma, dbeta = dwdb[:C], dwdb[C:]
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.ln_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     mean.data_ptr(), rstd.data_ptr(), dx.data_ptr(),
                     dgamma.data_ptr(), dbeta.data_ptr(), R, C, s)
        return (dx, dgamma.to(w.dtype),
                dbeta.to(w.dtype) if ctx.has_bias else None, None)


def fused_layer_norm(x: torch.Tensor, weight: torch.Tensor,
                     bias: torch.Tensor, eps: float) -> torch.Tensor:
    return _FusedLayerNormFn.apply(x, weight, bias, eps)


def can_use(x: torch.Tensor, weight: torch.Tensor) -> bool:
    return (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and x.shape[-1] == weight.numel() and x.shape[-1] <= MAX_C
            and x.shape[-1] % 2 == 0)
