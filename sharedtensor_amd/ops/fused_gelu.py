"""Fused bf16 GELU-tanh (hand-written CDNA4 kernels, csrc/gelu_kernels.hip).

Measured SLOWER than torch's gelu forward on MI355X (0.251 vs 0.187 ms at
(65536, 3072); backward ties) — torch's 8-wide vectorized elementwise wins.
Kept as a correct, tested alternative; NOT wired into the models."""
from __future__ import annotations

import torch

from .. import _core


class _FusedGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        y = torch.empty_like(x)
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.gelu_fwd(x.data_ptr(), y.data_ptr(), x.numel(), s)
        ctx.save_for_backward(x)
        return y

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.empty_like(x)
        s = torch.cuda.current_stream(x.device).cuda_stream
        _core.gelu_bwd(dy.data_ptr(), x.data_ptr(), dx.data_ptr(), x.numel(), s)
        return dx


def fused_gelu(x: torch.Tensor) -> torch.Tensor:
    return _FusedGeluFn.apply(x)


def can_use(x: torch.Tensor) -> bool:
    return x.is_cuda and x.dtype == torch.bfloat16 and x.numel() % 8 == 0
