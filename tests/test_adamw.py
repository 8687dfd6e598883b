"""Fused AdamW vs torch.optim.AdamW (CPU numerics; GPU variant in
tests/test_gpu_adamw.py).

The engine's AdamW (csrc/hip_kernels.hip k_fused_adamw / the CPU loop in
engine.cpp) must reproduce torch.optim.AdamW exactly on a master-only
engine: values == torch's parameter after K steps with identical grads."""
import torch

from sharedtensor_amd.engine import SharedFlat
from sharedtensor_amd.parallel.async_dp import AsyncAdamW, AsyncDPTrainer
from sharedtensor_amd.utils import free_port

LR, BETAS, EPS, WD = 2e-2, (0.9, 0.95), 1e-8, 0.01


def test_matches_torch_adamw_cpu():
    torch.manual_seed(3)
    n = 4099  # odd: exercises tail handling
    init = torch.randn(n)
    grads = [torch.randn(n) for _ in range(6)]

    # torch reference
    p = torch.nn.Parameter(init.clone())
    opt = torch.optim.AdamW([p], lr=LR, betas=BETAS, eps=EPS, weight_decay=WD)
    for g in grads:
        opt.zero_grad()
        p.grad = g.clone()
        opt.step()

    # engine (master only: values evolve exactly by the fused update)
    sh = SharedFlat("127.0.0.1", free_port(), [n], device="cpu",
                    provision_up=False, expected_children=0)
    sh._start()
    try:
        sh._add_flat(init.clone())
        mom = torch.zeros(n)
        vel = torch.zeros(n)
        for step, g in enumerate(grads, start=1):
            sh.fused_adamw_step(mom, vel, g, step, LR, BETAS, EPS, WD)
        torch.testing.assert_close(sh.values, p.detach(), rtol=1e-5,
                                   atol=1e-6)
    finally:
        sh.close()


def test_trainer_adamw_converges_cpu():
    from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
    torch.manual_seed(5)
    model = GPT2(GPT2Config.tiny())
    tr = AsyncDPTrainer(model, port_base=free_port(), rank=0, world=1,
                        lr=1e-3, optimizer="adamw", weight_decay=0.01,
                        amp_dtype=None)
    try:
        assert isinstance(tr.opt, AsyncAdamW)
        x = torch.randint(0, 256, (2, 33))
        losses = [float(tr.step(x[:, :-1], x[:, 1:])) for _ in range(8)]
        assert losses[-1] < losses[0], losses
        assert all(torch.isfinite(torch.tensor(losses)))
    finally:
        tr.close()
