"""GPT-2 model family — the flagship async-DP workload.

The reference repo names char-rnn as its intended training example
(/root/reference/README.md:37); BASELINE.json config 3 upgrades that to
GPT-2-small (char-rnn-style next-token loss) trained async-data-parallel with
all parameters as one shared tensor.  The model is plain PyTorch-ROCm: on
MI355X the matmuls run on MFMA via rocBLAS/hipBLASLt and attention via
torch SDPA; the shared-parameter machinery (this framework's contribution)
runs in our HIP kernels underneath it.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_layer: int = 12
    n_head: int = 12
    n_embd: int = 768
    block_size: int = 1024
    dropout: float = 0.0

    @classmethod
    def small(cls):  # 124M — the benchmark config
        return cls()

    @classmethod
    def tiny(cls):  # for CPU tests
        return cls(vocab_size=256, n_layer=2, n_head=2, n_embd=64,
                   block_size=64)


class FusedLayerNorm(nn.LayerNorm):
    """LayerNorm that stays in bf16 under autocast.

    torch's autocast policy runs LayerNorm in fp32, casting the whole
    (B, T, C) activation up and back down every call — measured at ~9% of a
    GPT-2-small step on MI355X (profiles/gpt2_train_step_kernels.txt).  The
    native layer_norm kernel already accumulates statistics in fp32, so when
    weights and input share a low-precision dtype we bypass the autocast
    upcast entirely.
    """

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16
                and self.weight.dtype == torch.bfloat16):
            from ..ops import fused_ln
            if fused_ln.can_use(x, self.weight):
                return fused_ln.fused_layer_norm(x, self.weight, self.bias,
                                                 self.eps)
            with torch.autocast("cuda", enabled=False):
                return F.layer_norm(x, self.normalized_shape, self.weight,
                                    self.bias, self.eps)
        return super().forward(x)


class CausalSelfAttention(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        assert cfg.n_embd % cfg.n_head == 0
        self.qkv = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.proj = nn.Linear(cfg.n_embd, cfg.n_embd)
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head

    def forward(self, x):
        B, T, C = x.shape
        q, k, v = self.qkv(x).split(C, dim=2)
        q = q.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        k = k.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        v = v.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        if x.is_cuda:
            # CK efficient-attention measured +16% end-to-end over the
            # default aotriton flash backend on MI355X at these shapes
            # (profiles/README.md), identical loss; priority list falls
            # back when a shape is unsupported
            from torch.nn.attention import SDPBackend, sdpa_kernel
            with sdpa_kernel([SDPBackend.EFFICIENT_ATTENTION,
                              SDPBackend.FLASH_ATTENTION, SDPBackend.MATH],
                             set_priority=True):
                y = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        else:
            y = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        y = y.transpose(1, 2).contiguous().view(B, T, C)
        return self.proj(y)


class MLP(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.fc = nn.Linear(cfg.n_embd, 4 * cfg.n_embd)
        self.proj = nn.Linear(4 * cfg.n_embd, cfg.n_embd)

    def forward(self, x):
        # torch's gelu measured FASTER than our fused kernel here (0.187 vs
        # 0.251 ms fwd on the B=64 shape; ops/fused_gelu.py kept as a
        # measured alternative) — keep the library kernel
        return self.proj(F.gelu(self.fc(x), approximate="tanh"))


class Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.ln1 = FusedLayerNorm(cfg.n_embd)
        self.attn = CausalSelfAttention(cfg)
        self.ln2 = FusedLayerNorm(cfg.n_embd)
        self.mlp = MLP(cfg)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        x = x + self.mlp(self.ln2(x))
        return x


class GPT2(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # weight tying
        self.apply(self._init)
        for name, p in self.named_parameters():
            if name.endswith("proj.weight"):
                nn.init.normal_(p, std=0.02 / math.sqrt(2 * cfg.n_layer))

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, idx, targets=None):
        B, T = idx.shape
        pos = torch.arange(T, device=idx.device)
        x = self.wte(idx) + self.wpe(pos)
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        if targets is None:
            return logits, None
        flat = logits.view(-1, logits.size(-1))
        tgt = targets.reshape(-1)
        if flat.is_cuda and flat.dtype == torch.bfloat16:
            from ..ops import fused_ce
            if fused_ce.can_use(flat):
                return logits, fused_ce.fused_cross_entropy(flat, tgt)
        loss = F.cross_entropy(flat, tgt)
        return logits, loss

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())
