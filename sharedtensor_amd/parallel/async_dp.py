"""Async data parallelism over the shared-tensor engine.

The reference's one parallelism strategy: N workers
each run a local training loop against a replicated parameter tensor; the
engine gossips compressed deltas continuously in the background, fully
overlapped with compute.  There is no lockstep all-reduce — staleness is
bounded by the compression scale, not by synchronization.

This module wires a torch model into that scheme MI355X-style:
  * all parameters live as views of one flat fp32 replica in HBM3E
    (zero-copy: the forward pass reads the live gossip target)
  * all gradients accumulate into one flat buffer (views as .grad)
  * the optimizer update is ONE fused HIP kernel (k_fused_sgd) that computes
    momentum, applies -lr*m to the replica AND stages it into every link's
    residual delta in a single HBM pass.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from ..engine import _SharedBase


def tree_parent(rank: int) -> int:
    """rank 0 is the root; children of r are 2r+1, 2r+2 (balanced binary
    tree over one node's GPUs — each edge is one xGMI hop)."""
    return (rank - 1) // 2


def tree_children(rank: int, world: int):
    return [c for c in (2 * rank + 1, 2 * rank + 2) if c < world]


class FlatParamShared(_SharedBase):
    """Shares a model's parameters as one flat fp32 tensor (BASELINE
    config 3: 'params as one shared tensor')."""

    def __init__(self, model: torch.nn.Module, host: str, port_base: int,
                 rank: int, world: int, param_dtype: torch.dtype = torch.float32,
                 **kw):
        params = [p for p in model.parameters() if p.requires_grad]
        # dedupe tied parameters (GPT-2 ties lm_head.weight to wte.weight)
        seen, uniq = set(), []
        for p in params:
            if id(p) not in seen:
                seen.add(id(p))
                uniq.append(p)
        self.params = uniq
        sizes = [p.numel() for p in self.params]
        device = self.params[0].device
        n = sum(sizes)

        nchild = len(tree_children(rank, world))
        explicit_parent = ""
        listen_port = 0
        if world > 1:
            listen_port = port_base + rank
            if rank > 0:
                explicit_parent = f"{host}:{port_base + tree_parent(rank)}"
        kw.setdefault("expected_children", nchild if world > 1 else 0)
        super().__init__(host, port_base, [n], device=device,
                         provision_up=rank > 0,
                         explicit_parent=explicit_parent,
                         listen_port=listen_port, **kw)

        # snapshot initial params, re-point them into the replica slab (fp32)
        # or into a bf16 shadow of it (mixed precision: bf16 matmuls without
        # autocast's per-op weight casts; fp32 master stays in the engine)
        init_flat = torch.cat([p.detach().reshape(-1) for p in self.params]).float()
        self.param_dtype = param_dtype
        self.mom_flat = torch.zeros(n, dtype=torch.float32, device=device)
        if param_dtype == torch.bfloat16:
            if not self._gpu:
                raise ValueError("bf16 shadow params need a GPU")
            self.shadow = torch.empty(n, dtype=torch.bfloat16, device=device)
            self.grad_flat = torch.zeros(n, dtype=torch.bfloat16, device=device)
            param_src = self.shadow
        else:
            self.shadow = None
            self.grad_flat = torch.zeros(n, dtype=torch.float32, device=device)
            param_src = self.values
        off = 0
        for p in self.params:
            sz = p.numel()
            p.data = param_src[off:off + sz].view(p.shape)
            p.grad = self.grad_flat[off:off + sz].view(p.shape)
            off += sz

        self._start()
        if self.is_master:
            # seed the shared state with this rank's init (master's weights
            # win; other ranks receive them via snapshot/gossip)
            self._add_flat(init_flat)
        if self.shadow is not None:
            # initial shadow refresh (post-join: values now hold the state)
            self.shadow.copy_(self.values)
        self.rank = rank
        self.world = world


class AsyncSGD:
    """SGD-momentum whose update feeds the shared tensor through the fused
    kernel — the optimizer step IS the addFromTensor."""

    def __init__(self, shared: FlatParamShared, lr: float = 0.1,
                 momentum: float = 0.9):
        self.shared = shared
        self.lr = lr
        self.momentum = momentum

    def step(self):
        if self.shared.shadow is not None:
            self.shared.fused_sgd_bf16_step(self.shared.mom_flat,
                                            self.shared.grad_flat,
                                            self.shared.shadow, self.lr,
                                            self.momentum)
        else:
            self.shared.fused_sgd_step(self.shared.mom_flat,
                                       self.shared.grad_flat,
                                       self.lr, self.momentum)

    def zero_grad(self, set_to_none: bool = False):
        self.shared.grad_flat.zero_()


class AsyncAdamW:
    """AdamW whose update feeds the shared tensor through the fused kernel
    (torch.optim.AdamW semantics: decoupled weight decay on the pre-update
    master weight) — the optimizer step IS the addFromTensor."""

    def __init__(self, shared: FlatParamShared, lr: float = 3e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.01):
        self.shared = shared
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.vel_flat = torch.zeros(shared.n, dtype=torch.float32,
                                    device=shared.device)

    def step(self):
        self.step_count += 1
        sh = self.shared
        if sh.shadow is not None:
            sh.fused_adamw_bf16_step(sh.mom_flat, self.vel_flat, sh.grad_flat,
                                     sh.shadow, self.step_count, self.lr,
                                     self.betas, self.eps, self.weight_decay)
        else:
            sh.fused_adamw_step(sh.mom_flat, self.vel_flat, sh.grad_flat,
                                self.step_count, self.lr, self.betas,
                                self.eps, self.weight_decay)

    def zero_grad(self, set_to_none: bool = False):
        self.shared.grad_flat.zero_()


class AsyncDPTrainer:
    """One rank's training loop against the shared parameter tensor."""

    def __init__(self, model: torch.nn.Module, host: str = "127.0.0.1",
                 port_base: Optional[int] = None, rank: Optional[int] = None,
                 world: Optional[int] = None, lr: float = 0.1,
                 momentum: float = 0.9, amp_dtype: Optional[torch.dtype] = torch.bfloat16,
                 param_dtype: Optional[torch.dtype] = None,
                 optimizer: str = "sgd", betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0, **engine_kw):
        rank = int(os.environ.get("RANK", 0)) if rank is None else rank
        world = int(os.environ.get("WORLD_SIZE", 1)) if world is None else world
        if port_base is None:
            port_base = int(os.environ.get("SHTENS_PORT_BASE", 21000))
        self.model = model
        if param_dtype is None:
            param_dtype = torch.float32
        self.shared = FlatParamShared(model, host, port_base, rank, world,
                                      param_dtype=param_dtype, **engine_kw)
        if optimizer == "adamw":
            self.opt = AsyncAdamW(self.shared, lr=lr, betas=betas, eps=eps,
                                  weight_decay=weight_decay)
        elif optimizer == "sgd":
            self.opt = AsyncSGD(self.shared, lr=lr, momentum=momentum)
        else:
            raise ValueError(f"unknown optimizer {optimizer!r}")
        self.amp_dtype = amp_dtype
        self.rank, self.world = rank, world
        self.device = self.shared.device

    def step(self, batch, targets) -> torch.Tensor:
        self.opt.zero_grad()
        use_amp = self.amp_dtype is not None and self.device.type == "cuda"
        with torch.autocast(device_type="cuda", dtype=self.amp_dtype,
                            enabled=use_amp):
            _, loss = self.model(batch, targets)
        loss.backward()
        self.opt.step()
        return loss.detach()

    def stats(self):
        return self.shared.stats()

    def close(self):
        self.shared.close()
