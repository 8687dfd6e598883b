"""GPT-2 model family unit tests (CPU)."""
import torch

from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config


def test_tiny_forward_backward():
    torch.manual_seed(0)
    cfg = GPT2Config.tiny()
    m = GPT2(cfg)
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    logits, loss = m(x[:, :-1], x[:, 1:])
    assert logits.shape == (2, 15, cfg.vocab_size)
    assert torch.isfinite(loss)
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())


def test_small_param_count():
    cfg = GPT2Config.small()
    m = GPT2(cfg)
    n = m.num_params()
    # GPT-2-small: ~124M (163M counting the tied head twice; num_params
    # iterates parameters() which includes wte once due to tying)
    uniq = sum(p.numel() for p in {id(p): p for p in m.parameters()}.values())
    assert 120e6 < uniq < 130e6, uniq


def test_loss_decreases_on_overfit():
    torch.manual_seed(1)
    cfg = GPT2Config.tiny()
    m = GPT2(cfg)
    x = torch.randint(0, cfg.vocab_size, (1, 32))
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        _, loss = m(x[:, :-1], x[:, 1:])
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, losses[::10]
