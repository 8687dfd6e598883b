"""Fused bf16 SwiGLU (csrc/swiglu_kernels.hip): y = silu(gate) * up in one
kernel per direction — torch runs silu and mul separately (5 activation
passes forward, 8 backward vs our 3/5) on the (B*T, ffn_dim) tensors of the
Llama MLP."""
from __future__ import annotations

import os

import torch

from .. import _core


class _FusedSwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x1, x3):
        x1 = x1.contiguous()
        x3 = x3.contiguous()
        y = torch.empty_like(x1)
        s = torch.cuda.current_stream(x1.device).cuda_stream
        _core.swiglu_fwd(x1.data_ptr(), x3.data_ptr(), y.data_ptr(),
                         x1.numel(), s)
        ctx.save_for_backward(x1, x3)
        return y

    @staticmethod
    def backward(ctx, dy):
        x1, x3 = ctx.saved_tensors
        dy = dy.contiguous()
        dx1 = torch.empty_like(x1)
        dx3 = torch.empty_like(x3)
        s = torch.cuda.current_stream(x1.device).cuda_stream
        _core.swiglu_bwd(dy.data_ptr(), x1.data_ptr(), x3.data_ptr(),
                         dx1.data_ptr(), dx3.data_ptr(), x1.numel(), s)
        return dx1, dx3


def fused_swiglu(x1: torch.Tensor, x3: torch.Tensor) -> torch.Tensor:
    return _FusedSwiGLUFn.apply(x1, x3)


def can_use(x1: torch.Tensor, x3: torch.Tensor) -> bool:
    if os.environ.get("SHTENS_NO_FUSED_SWIGLU") == "1":  # A/B knob
        return False
    return (x1.is_cuda and x1.dtype == torch.bfloat16
            and x3.dtype == torch.bfloat16 and x1.shape == x3.shape
            and x1.numel() % 8 == 0)
