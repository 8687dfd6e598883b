"""Fused AdamW HIP kernel vs torch.optim.AdamW on GPU (fp32 and the
bf16-grad + bf16-shadow mixed-precision variant)."""
import pytest
import torch

from sharedtensor_amd.engine import SharedFlat
from sharedtensor_amd.utils import free_port

pytestmark = pytest.mark.gpu

LR, BETAS, EPS, WD = 2e-2, (0.9, 0.95), 1e-8, 0.01
N = (1 << 20) + 3


def _torch_ref(init, grads):
    p = torch.nn.Parameter(init.clone())
    opt = torch.optim.AdamW([p], lr=LR, betas=BETAS, eps=EPS, weight_decay=WD)
    for g in grads:
        opt.zero_grad()
        p.grad = g.clone().float()
        opt.step()
    return p.detach()


def test_fp32_matches_torch():
    torch.cuda.set_device(0)
    torch.manual_seed(11)
    init = torch.randn(N, device="cuda")
    grads = [torch.randn(N, device="cuda") for _ in range(5)]
    ref = _torch_ref(init, grads)
    sh = SharedFlat("127.0.0.1", free_port(), [N], device="cuda:0",
                    provision_up=False, expected_children=0)
    sh._start()
    try:
        sh._add_flat(init.clone())
        mom = torch.zeros(N, device="cuda")
        vel = torch.zeros(N, device="cuda")
        for step, g in enumerate(grads, start=1):
            sh.fused_adamw_step(mom, vel, g, step, LR, BETAS, EPS, WD)
        torch.cuda.synchronize()
        torch.testing.assert_close(sh.values, ref, rtol=1e-5, atol=1e-6)
    finally:
        sh.close()


def test_bf16_grads_shadow():
    torch.cuda.set_device(0)
    torch.manual_seed(12)
    init = torch.randn(N, device="cuda")
    grads16 = [torch.randn(N, device="cuda").to(torch.bfloat16)
               for _ in range(5)]
    ref = _torch_ref(init, grads16)  # fp32 master fed the same bf16 grads
    sh = SharedFlat("127.0.0.1", free_port(), [N], device="cuda:0",
                    provision_up=False, expected_children=0)
    sh._start()
    try:
        sh._add_flat(init.clone())
        mom = torch.zeros(N, device="cuda")
        vel = torch.zeros(N, device="cuda")
        shadow = torch.zeros(N, dtype=torch.bfloat16, device="cuda")
        for step, g in enumerate(grads16, start=1):
            sh.fused_adamw_bf16_step(mom, vel, g, shadow, step, LR, BETAS,
                                     EPS, WD)
        torch.cuda.synchronize()
        torch.testing.assert_close(sh.values, ref, rtol=1e-5, atol=1e-6)
        # shadow is the bf16 rounding of the fp32 master
        torch.testing.assert_close(shadow.float(), sh.values, rtol=1e-2,
                                   atol=1e-2)
    finally:
        sh.close()
