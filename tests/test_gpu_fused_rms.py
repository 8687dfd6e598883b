"""Fused bf16 RMSNorm kernels vs a plain fp32 torch reference."""
import pytest
import torch

from sharedtensor_amd.ops import fused_rms

pytestmark = pytest.mark.gpu


def ref_rms(x32, w32, eps):
    h = x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + eps)
    return h * w32


@pytest.mark.parametrize("R,C", [(128, 2048), (1000, 2048), (64, 4096),
                                 (32, 64), (7, 1536)])
def test_fwd_bwd_matches_fp32_torch(R, C):
    torch.manual_seed(R + C)
    eps = 1e-5
    x = (torch.randn(R, C, device="cuda") * 2).to(torch.bfloat16)
    w = torch.randn(C, device="cuda").to(torch.bfloat16)
    dy = torch.randn(R, C, device="cuda").to(torch.bfloat16)

    # plain fp32 torch reference of the same op
    xt = x.float().clone().requires_grad_(True)
    wt = w.float().clone().requires_grad_(True)
    yt = ref_rms(xt, wt, eps)
    yt.backward(dy.float())

    xf = x.clone().requires_grad_(True)
    wf = w.clone().requires_grad_(True)
    yf = fused_rms.fused_rms_norm(xf, wf, eps)
    yf.backward(dy)
    torch.cuda.synchronize()

    torch.testing.assert_close(yf.float(), yt.float(), rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(xf.grad.float(), xt.grad.float(), rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(wf.grad.float(), wt.grad.float(), rtol=3e-2,
                               atol=3e-1)


def test_model_path_uses_fused_kernel():
    from sharedtensor_amd.models.llama import RMSNorm
    torch.manual_seed(0)
    rn = RMSNorm(2048, 1e-5).cuda().to(torch.bfloat16)
    x = torch.randn(4, 32, 2048, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    assert fused_rms.can_use(x, rn.weight)
    y = rn(x)
    assert y.dtype == torch.bfloat16 and y.shape == x.shape
    y.sum().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(x.grad.float()).all()
    assert rn.weight.grad is not None
    assert torch.isfinite(rn.weight.grad.float()).all()


def test_training_equivalence_small_llama():
    """A few training steps with fused vs module RMSNorm must track."""
    from sharedtensor_amd.models.llama import Llama, LlamaConfig
    import sharedtensor_amd.ops.fused_rms as fr
    losses = {}
    for tag in ("fused", "torch"):
        torch.manual_seed(7)
        cfg = LlamaConfig.tiny()
        m = Llama(cfg).cuda().to(torch.bfloat16)
        opt = torch.optim.SGD(m.parameters(), lr=0.05)
        x = torch.randint(0, cfg.vocab_size, (2, 33), device="cuda")
        ls = []
        orig = fr.can_use
        if tag == "torch":
            fr.can_use = lambda *a: False
        try:
            for _ in range(8):
                opt.zero_grad()
                _, loss = m(x[:, :-1], x[:, 1:])
                loss.float().backward()
                opt.step()
                ls.append(float(loss.detach()))
        finally:
            fr.can_use = orig
        losses[tag] = ls
    for a, b in zip(losses["fused"], losses["torch"]):
        assert abs(a - b) < 0.15, (losses["fused"], losses["torch"])
