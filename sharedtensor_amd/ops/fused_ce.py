"""Fused bf16 cross-entropy (hand-written CDNA4 kernels, csrc/ce_kernels.hip).

Replaces torch's softmax-forward + softmax-backward pair (and its fp32
intermediates) for the (B*T, vocab) bf16 logits of the language-model head:
one online-logsumexp pass forward, one dlogits pass backward, fp32 math,
bf16 tensors.  reduction='mean' semantics.
"""
from __future__ import annotations

import torch

from .. import _core


class _FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        logits = logits.contiguous()
        R, V = logits.shape
        t32 = targets.to(torch.int32).contiguous()
        loss = torch.empty(R, dtype=torch.float32, device=logits.device)
        row_m = torch.empty(R, dtype=torch.float32, device=logits.device)
        row_lse = torch.empty(R, dtype=torch.float32, device=logits.device)
        s = torch.cuda.current_stream(logits.device).cuda_stream
        _core.ce_fwd(logits.data_ptr(), t32.data_ptr(), loss.data_ptr(),
                     row_m.data_ptr(), row_lse.data_ptr(), R, V, s)
        ctx.save_for_backward(logits, t32, row_lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, dloss):
        logits, t32, row_lse = ctx.saved_tensors
        R, V = logits.shape
        dlogits = torch.empty_like(logits)
        g = dloss.to(device=logits.device, dtype=torch.float32).contiguous()
        s = torch.cuda.current_stream(logits.device).cuda_stream
        _core.ce_bwd(logits.data_ptr(), t32.data_ptr(), row_lse.data_ptr(),
                     dlogits.data_ptr(), g.data_ptr(), 1.0 / R, R, V, s)
        return dlogits, None


def fused_cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over rows; logits (R, V) bf16 cuda, targets (R,)."""
    return _FusedCEFn.apply(logits, targets)


def can_use(logits: torch.Tensor) -> bool:
    return (logits.is_cuda and logits.dtype == torch.bfloat16
            and logits.dim() == 2)
