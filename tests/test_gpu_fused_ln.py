"""Fused bf16 LayerNorm kernels vs torch's native layer_norm."""
import pytest
import torch

from sharedtensor_amd.ops import fused_ln

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("R,C", [(128, 768), (1000, 768), (64, 1536),
                                 (32, 64), (7, 4096)])
def test_fwd_bwd_matches_torch(R, C):
    torch.manual_seed(R + C)
    x = (torch.randn(R, C, device="cuda") * 2).to(torch.bfloat16)
    w = torch.randn(C, device="cuda").to(torch.bfloat16)
    b = torch.randn(C, device="cuda").to(torch.bfloat16)
    dy = torch.randn(R, C, device="cuda").to(torch.bfloat16)

    # torch reference (bf16 in/out, fp32 internals — same contract)
    xt = x.clone().requires_grad_(True)
    wt = w.clone().requires_grad_(True)
    bt = b.clone().requires_grad_(True)
    yt = torch.nn.functional.layer_norm(xt, (C,), wt, bt, 1e-5)
    yt.backward(dy)

    xf = x.clone().requires_grad_(True)
    wf = w.clone().requires_grad_(True)
    bf = b.clone().requires_grad_(True)
    yf = fused_ln.fused_layer_norm(xf, wf, bf, 1e-5)
    yf.backward(dy)
    torch.cuda.synchronize()

    torch.testing.assert_close(yf.float(), yt.float(), rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(xf.grad.float(), xt.grad.float(), rtol=5e-2,
                               atol=5e-2)
    torch.testing.assert_close(wf.grad.float(), wt.grad.float(), rtol=3e-2,
                               atol=3e-1)
    torch.testing.assert_close(bf.grad.float(), bt.grad.float(), rtol=3e-2,
                               atol=3e-1)


def test_3d_input_and_model_path():
    from sharedtensor_amd.models.gpt2 import FusedLayerNorm
    torch.manual_seed(0)
    ln = FusedLayerNorm(768).cuda().to(torch.bfloat16)
    x = torch.randn(4, 32, 768, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ln(x)
    assert y.dtype == torch.bfloat16 and y.shape == x.shape
    y.sum().backward()
    torch.cuda.synchronize()
    assert torch.isfinite(x.grad.float()).all()
    assert ln.weight.grad is not None and torch.isfinite(ln.weight.grad.float()).all()


def test_training_equivalence_small_model():
    """A few training steps with fused vs torch LN must track closely."""
    from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
    losses = {}
    for tag in ("fused", "torch"):
        torch.manual_seed(7)
        cfg = GPT2Config.tiny()
        m = GPT2(cfg).cuda().to(torch.bfloat16)
        opt = torch.optim.SGD(m.parameters(), lr=0.05)
        x = torch.randint(0, cfg.vocab_size, (2, 33), device="cuda")
        ls = []
        import sharedtensor_amd.ops.fused_ln as fl
        orig = fl.can_use
        if tag == "torch":
            fl.can_use = lambda *a: False
        try:
            for _ in range(8):
                opt.zero_grad()
                _, loss = m(x[:, :-1], x[:, 1:])
                loss.float().backward()
                opt.step()
                ls.append(float(loss.detach()))
        finally:
            fl.can_use = orig
        losses[tag] = ls
    for a, b in zip(losses["fused"], losses["torch"]):
        assert abs(a - b) < 0.15, (losses["fused"], losses["torch"])
