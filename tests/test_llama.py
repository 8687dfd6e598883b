"""Llama model family unit tests (CPU)."""
import torch

from sharedtensor_amd.models.llama import Llama, LlamaConfig


def test_tiny_forward_backward():
    torch.manual_seed(0)
    cfg = LlamaConfig.tiny()
    m = Llama(cfg)
    x = torch.randint(0, cfg.vocab_size, (2, 16))
    logits, loss = m(x[:, :-1], x[:, 1:])
    assert logits.shape == (2, 15, cfg.vocab_size)
    assert torch.isfinite(loss)
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())


def test_gqa_heads():
    cfg = LlamaConfig.tiny()
    assert cfg.n_head != cfg.n_kv_head  # GQA is actually exercised
    m = Llama(cfg)
    x = torch.randint(0, cfg.vocab_size, (1, 8))
    logits, _ = m(x)
    assert torch.isfinite(logits).all()


def test_llama3_8b_config_shape():
    cfg = LlamaConfig.llama3_8b()
    # parameter count ~8B without instantiating the model
    d, f, L, V = cfg.dim, cfg.ffn_dim, cfg.n_layer, cfg.vocab_size
    hd = d // cfg.n_head
    attn = d * d + 2 * d * (cfg.n_kv_head * hd) + d * d
    mlp = 3 * d * f
    total = L * (attn + mlp + 2 * d) + 2 * V * d + d
    assert 7.5e9 < total < 8.5e9, total


def test_overfit_tiny():
    torch.manual_seed(3)
    cfg = LlamaConfig.tiny()
    m = Llama(cfg)
    x = torch.randint(0, cfg.vocab_size, (1, 32))
    opt = torch.optim.SGD(m.parameters(), lr=0.05, momentum=0.9)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        _, loss = m(x[:, :-1], x[:, 1:])
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.8, losses[::10]


def test_rope_rotation_norm_preserving():
    from sharedtensor_amd.models.llama import apply_rope, rope_freqs
    cos, sin = rope_freqs(16, 32, 10000.0, "cpu")
    x = torch.randn(2, 4, 32, 16)
    y = apply_rope(x, cos, sin)
    # rotations preserve the norm of each (even, odd) pair
    torch.testing.assert_close(x.norm(dim=-1), y.norm(dim=-1), rtol=1e-5,
                               atol=1e-5)
    # position 0 is the identity rotation
    torch.testing.assert_close(y[:, :, 0], x[:, :, 0])
