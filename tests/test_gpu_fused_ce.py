"""Fused bf16 cross-entropy kernels vs torch reference."""
import pytest
import torch

from sharedtensor_amd.ops import fused_ce

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("R,V", [(64, 1000), (256, 50257), (7, 13)])
def test_fwd_bwd_matches_torch(R, V):
    torch.manual_seed(R + V)
    logits = (torch.randn(R, V, device="cuda") * 3).to(torch.bfloat16)
    targets = torch.randint(0, V, (R,), device="cuda")

    lt = logits.clone().requires_grad_(True)
    loss_t = torch.nn.functional.cross_entropy(lt.float(), targets)
    loss_t.backward()

    lf = logits.clone().requires_grad_(True)
    loss_f = fused_ce.fused_cross_entropy(lf, targets)
    loss_f.backward()
    torch.cuda.synchronize()

    torch.testing.assert_close(loss_f.float(), loss_t.float(), rtol=2e-3,
                               atol=2e-3)
    torch.testing.assert_close(lf.grad.float(), lt.grad.float(), rtol=5e-2,
                               atol=1e-4)


def test_upstream_grad_scaling():
    torch.manual_seed(1)
    R, V = 32, 512
    logits = torch.randn(R, V, device="cuda").to(torch.bfloat16)
    targets = torch.randint(0, V, (R,), device="cuda")
    lf = logits.clone().requires_grad_(True)
    loss = fused_ce.fused_cross_entropy(lf, targets)
    (loss * 3.0).backward()
    lf2 = logits.clone().requires_grad_(True)
    loss2 = fused_ce.fused_cross_entropy(lf2, targets)
    loss2.backward()
    torch.cuda.synchronize()
    torch.testing.assert_close(lf.grad.float(), 3.0 * lf2.grad.float(),
                               rtol=2e-2, atol=1e-5)


def test_model_path_and_training():
    from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
    torch.manual_seed(3)
    cfg = GPT2Config.tiny()
    m = GPT2(cfg).cuda().to(torch.bfloat16)
    opt = torch.optim.SGD(m.parameters(), lr=0.05)
    x = torch.randint(0, cfg.vocab_size, (2, 33), device="cuda")
    losses = []
    for _ in range(12):
        opt.zero_grad()
        _, loss = m(x[:, :-1], x[:, 1:])
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses
