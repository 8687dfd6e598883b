"""Fused bf16 SwiGLU kernels vs a plain fp32 torch reference."""
import pytest
import torch
import torch.nn.functional as F

from sharedtensor_amd.ops import fused_swiglu

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("shape", [(128, 8192), (64, 14336), (3, 128)])
def test_fwd_bwd_matches_fp32_torch(shape):
    torch.manual_seed(sum(shape))
    x1 = (torch.randn(shape, device="cuda") * 2).to(torch.bfloat16)
    x3 = torch.randn(shape, device="cuda").to(torch.bfloat16)
    dy = torch.randn(shape, device="cuda").to(torch.bfloat16)

    a = x1.float().clone().requires_grad_(True)
    b = x3.float().clone().requires_grad_(True)
    yt = F.silu(a) * b
    yt.backward(dy.float())

    af = x1.clone().requires_grad_(True)
    bf = x3.clone().requires_grad_(True)
    assert fused_swiglu.can_use(af, bf)
    yf = fused_swiglu.fused_swiglu(af, bf)
    yf.backward(dy)
    torch.cuda.synchronize()

    torch.testing.assert_close(yf.float(), yt.float(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(af.grad.float(), a.grad, rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(bf.grad.float(), b.grad, rtol=5e-2, atol=5e-2)


def test_model_path_and_equivalence():
    from sharedtensor_amd.models.llama import Llama, LlamaConfig
    import sharedtensor_amd.ops.fused_swiglu as fs
    losses = {}
    for tag in ("fused", "torch"):
        torch.manual_seed(9)
        cfg = LlamaConfig.tiny()
        m = Llama(cfg).cuda().to(torch.bfloat16)
        opt = torch.optim.SGD(m.parameters(), lr=0.05)
        x = torch.randint(0, cfg.vocab_size, (2, 33), device="cuda")
        orig = fs.can_use
        if tag == "torch":
            fs.can_use = lambda *a: False
        ls = []
        try:
            for _ in range(8):
                opt.zero_grad()
                _, loss = m(x[:, :-1], x[:, 1:])
                loss.float().backward()
                opt.step()
                ls.append(float(loss.detach()))
        finally:
            fs.can_use = orig
        losses[tag] = ls
    for a, b in zip(losses["fused"], losses["torch"]):
        assert abs(a - b) < 0.15, (losses["fused"], losses["torch"])
