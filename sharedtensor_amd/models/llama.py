"""Llama-3-family models (RMSNorm, RoPE, GQA, SwiGLU).

BASELINE.json config 4 shares a Llama-3-8B's parameters as a
table-of-tensors (`SharedTable`, one scale per tensor) across 8 MI355X.
Llama-3-8B in fp32 is ~32 GB of parameters; replica + up to 3 link residual
deltas = <=128 GB, comfortably inside 288 GB of HBM3E per GPU.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    n_layer: int = 32
    n_head: int = 32
    n_kv_head: int = 8
    dim: int = 4096
    ffn_dim: int = 14336
    block_size: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5

    @classmethod
    def llama3_8b(cls):
        return cls()

    @classmethod
    def llama_1b(cls):  # Llama-3.2-1B-ish, for single-GPU runs
        return cls(n_layer=16, n_head=32, n_kv_head=8, dim=2048,
                   ffn_dim=8192, block_size=4096)

    @classmethod
    def tiny(cls):  # for CPU tests
        return cls(vocab_size=256, n_layer=2, n_head=4, n_kv_head=2, dim=64,
                   ffn_dim=128, block_size=64)


class RMSNorm(nn.Module):
    def __init__(self, dim, eps):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        if x.is_cuda and x.dtype == torch.bfloat16 \
                and self.weight.dtype == torch.bfloat16:
            # fused CDNA4 kernel: no fp32 upcast traffic on the (B,T,C)
            # activation (csrc/ln_kernels.hip, fp32 statistics inside)
            from ..ops import fused_rms
            if fused_rms.can_use(x, self.weight):
                return fused_rms.fused_rms_norm(x, self.weight, self.eps)
        dt = x.dtype
        x = x.float()
        x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return (x * self.weight.float()).to(dt)


def rope_freqs(head_dim, max_seq, theta, device):
    inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device).float() / head_dim))
    t = torch.arange(max_seq, device=device).float()
    f = torch.outer(t, inv)
    return torch.cos(f), torch.sin(f)


def apply_rope(x, cos, sin):
    # x: (B, H, T, D); pairwise rotate [x0, x1] halves interleaved as (even, odd).
    # Rotation runs in fp32 (cos/sin cache dtype); the result returns to
    # x's dtype so q/k match v for SDPA outside autocast.
    T = x.shape[2]
    c, s = cos[:T], sin[:T]  # (T, D/2)
    x1, x2 = x[..., 0::2], x[..., 1::2]
    o1 = x1 * c - x2 * s
    o2 = x1 * s + x2 * c
    out = torch.stack((o1, o2), dim=-1).flatten(-2)
    return out.to(x.dtype)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.head_dim = cfg.dim // cfg.n_head
        self.n_head, self.n_kv = cfg.n_head, cfg.n_kv_head
        self.wq = nn.Linear(cfg.dim, cfg.n_head * self.head_dim, bias=False)
        self.wk = nn.Linear(cfg.dim, cfg.n_kv_head * self.head_dim, bias=False)
        self.wv = nn.Linear(cfg.dim, cfg.n_kv_head * self.head_dim, bias=False)
        self.wo = nn.Linear(cfg.n_head * self.head_dim, cfg.dim, bias=False)

    def forward(self, x, cos, sin):
        B, T, _ = x.shape
        q = self.wq(x).view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        k = self.wk(x).view(B, T, self.n_kv, self.head_dim).transpose(1, 2)
        v = self.wv(x).view(B, T, self.n_kv, self.head_dim).transpose(1, 2)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        if x.is_cuda:
            # same backend priority as GPT-2 (+16% measured; MATH fallback
            # covers shapes a backend rejects, e.g. GQA corner cases)
            from torch.nn.attention import SDPBackend, sdpa_kernel
            with sdpa_kernel([SDPBackend.EFFICIENT_ATTENTION,
                              SDPBackend.FLASH_ATTENTION, SDPBackend.MATH],
                             set_priority=True):
                y = F.scaled_dot_product_attention(
                    q, k, v, is_causal=True,
                    enable_gqa=self.n_kv != self.n_head)
        else:
            y = F.scaled_dot_product_attention(
                q, k, v, is_causal=True, enable_gqa=self.n_kv != self.n_head)
        y = y.transpose(1, 2).contiguous().view(B, T, -1)
        return self.wo(y)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.w1 = nn.Linear(cfg.dim, cfg.ffn_dim, bias=False)  # gate
        self.w3 = nn.Linear(cfg.dim, cfg.ffn_dim, bias=False)  # up
        self.w2 = nn.Linear(cfg.ffn_dim, cfg.dim, bias=False)  # down

    def forward(self, x):
        x1 = self.w1(x)
        x3 = self.w3(x)
        if x1.is_cuda and x1.dtype == torch.bfloat16:
            # one fused CDNA4 kernel instead of separate silu + mul
            # (csrc/swiglu_kernels.hip)
            from ..ops import fused_swiglu
            if fused_swiglu.can_use(x1, x3):
                return self.w2(fused_swiglu.fused_swiglu(x1, x3))
        return self.w2(F.silu(x1) * x3)


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = LlamaAttention(cfg)
        self.mlp_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x, cos, sin):
        x = x + self.attn(self.attn_norm(x), cos, sin)
        x = x + self.mlp(self.mlp_norm(x))
        return x


class Llama(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.tok = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.blocks = nn.ModuleList(LlamaBlock(cfg) for _ in range(cfg.n_layer))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        std = 0.02
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, std=std / (math.sqrt(2 * cfg.n_layer)
                                              if name.endswith(("wo.weight", "w2.weight")) else 1.0))
        self._rope = None

    def _rope_cache(self, device, dtype):
        if self._rope is None or self._rope[0].device != device:
            cos, sin = rope_freqs(self.cfg.dim // self.cfg.n_head,
                                  self.cfg.block_size, self.cfg.rope_theta,
                                  device)
            self._rope = (cos, sin)
        return self._rope

    def forward(self, idx, targets=None):
        x = self.tok(idx)
        cos, sin = self._rope_cache(idx.device, x.dtype)
        for blk in self.blocks:
            x = blk(x, cos, sin)
        x = self.norm(x)
        logits = self.lm_head(x)
        if targets is None:
            return logits, None
        loss = F.cross_entropy(logits.float().view(-1, logits.size(-1)),
                               targets.reshape(-1))
        return logits, loss

    def num_params(self):
        return sum(p.numel() for p in self.parameters())
