"""Unit tests for the codec oracle (sharedtensor_amd/ops/oracle.py).

Invariants verified here mirror the reference protocol
(/root/reference/src/sharedtensor.c:106-177):
  * power-of-two scale
  * exact error feedback: residual_after + decoded(sent) == residual_before
    in fp32, element-for-element
  * LSB-first bit layout identical to the reference's byte stream
"""
import math

import numpy as np
import pytest
import torch

from sharedtensor_amd.ops import oracle as oc


def ref_quantize_1bit(delta: np.ndarray, scale: float):
    """Literal (slow) transcription of the reference semantics
    (sharedtensor.c:166-174), used as a second, independent oracle."""
    d = delta.astype(np.float32).copy()
    n = d.size
    buf = np.zeros((n + 7) // 8, dtype=np.uint8)
    s = np.float32(scale)
    for i in range(n):
        if d[i] > 0:
            d[i] -= s
        else:
            buf[i // 8] |= 1 << (i % 8)
            d[i] += s
    return buf, d


def ref_save_deltas(buf: np.ndarray, n: int, scale: float):
    """sharedtensor.c:106-111."""
    s = np.float32(scale)
    out = np.zeros(n, dtype=np.float32)
    for i in range(n):
        bit = (buf[i // 8] >> (i % 8)) & 1
        out[i] = s - bit * 2 * s
    return out


def is_pow2(x):
    if x == 0:
        return True
    m, _ = math.frexp(x)
    return m == 0.5


@pytest.mark.parametrize("n", [1, 7, 64, 100, 1000, 4 * 5 * 6 * 2])
def test_1bit_matches_reference_bytes(n):
    rng = torch.Generator().manual_seed(n)
    d = torch.randn(n, generator=rng) * 3.0
    scale, payload, new_d = oc.encode_1bit(d)
    assert is_pow2(scale) and scale > 0
    ref_buf, ref_d = ref_quantize_1bit(d.numpy(), scale)
    assert payload[: ref_buf.size] == ref_buf.tobytes()
    np.testing.assert_array_equal(new_d.numpy(), ref_d)
    # decode parity with the reference's save_deltas
    dec = oc.decode_1bit(payload, scale, n)
    np.testing.assert_array_equal(dec.numpy(), ref_save_deltas(ref_buf, n, scale))


@pytest.mark.parametrize("codec", [oc.CODEC_1BIT, oc.CODEC_FP8, oc.CODEC_INT4])
@pytest.mark.parametrize("n", [1, 63, 64, 65, 1000])
def test_error_feedback_exact(codec, n):
    rng = torch.Generator().manual_seed(1234 + codec * 10 + n)
    d = torch.randn(n, generator=rng) * (10.0 ** torch.randint(-3, 4, (n,), generator=rng).float())
    scale, payload, new_d = oc.encode(codec, d)
    assert is_pow2(scale)
    sent = oc.decode(codec, payload, scale, n)
    # error feedback is the exact fp32 subtraction d - sent (the fp32 rounding
    # of that subtraction is part of the protocol, as in sharedtensor.c:169-173)
    np.testing.assert_array_equal(new_d.numpy(), (d - sent).numpy())
    # and conservation holds up to that single fp32 rounding
    err = (new_d + sent - d).abs().numpy()
    tol = np.maximum(np.abs(sent.numpy()), np.abs(d.numpy())) * 2 ** -23
    assert np.all(err <= tol + 1e-38)
    assert len(payload) == oc.payload_bytes(codec, n)
    assert len(payload) % 8 == 0


@pytest.mark.parametrize("codec", [oc.CODEC_1BIT, oc.CODEC_FP8, oc.CODEC_INT4])
def test_zero_residual_keepalive(codec):
    d = torch.zeros(100)
    scale, payload, new_d = oc.encode(codec, d)
    assert scale == 0.0
    assert payload == b"\x00" * len(payload)
    assert torch.all(new_d == 0)
    assert torch.all(oc.decode(codec, payload, scale, 100) == 0)


def test_fp8_bounded_error():
    d = torch.randn(4096) * 5
    scale, payload, new_d = oc.encode_fp8(d)
    # e4m3 relative error <= 2^-4 of value + scale quantum
    assert new_d.abs().max().item() <= max(d.abs().max().item() / 16.0, scale) + 1e-6


def test_int4_bounded_error():
    d = torch.randn(4096) * 5
    scale, payload, new_d = oc.encode_int4(d)
    assert new_d.abs().max().item() <= scale / 2 + 1e-6  # rounding to nearest level


def test_int4_roundtrip_levels():
    scale = 0.5
    q = torch.arange(-7, 8).float()
    d = q * scale
    s, payload, new_d = oc.encode_int4(d, scale)
    dec = oc.decode_int4(payload, s, d.numel())
    np.testing.assert_allclose(dec.numpy(), d.numpy(), atol=0)
    assert torch.all(new_d == 0)


def test_1bit_convergence_single_link():
    """Repeated encode rounds shrink the residual: after enough rounds the
    receiver's accumulated value approaches the original delta."""
    torch.manual_seed(0)
    d = torch.randn(512) * 2
    target = d.clone()
    acc = torch.zeros(512)
    for _ in range(200):
        scale, payload, d = oc.encode_1bit(d)
        if scale == 0:
            break
        acc += oc.decode_1bit(payload, scale, 512)
    assert torch.norm(acc - target) / torch.norm(target) < 1e-3


def test_pow2_helpers():
    assert oc.pow2_floor(1.0) == 1.0
    assert oc.pow2_floor(1.5) == 1.0
    assert oc.pow2_floor(2.0) == 2.0
    assert oc.pow2_floor(0.75) == 0.5
    assert oc.pow2_floor(0.0) == 0.0
    assert oc.pow2_ceil(1.0) == 1.0
    assert oc.pow2_ceil(1.5) == 2.0
    assert oc.pow2_ceil(0.75) == 1.0


def test_table_roundtrip():
    torch.manual_seed(3)
    sizes = [5, 64, 129, 1]
    deltas = [torch.randn(s) * (10 ** i) for i, s in enumerate(sizes)]
    for codec in (oc.CODEC_1BIT, oc.CODEC_FP8, oc.CODEC_INT4):
        ins = [d.clone() for d in deltas]
        scales, payload, residuals = oc.encode_table(codec, ins)
        decs = oc.decode_table(codec, payload, scales, sizes)
        for d, r, a in zip(deltas, residuals, decs):
            np.testing.assert_array_equal(r.numpy(), (d - a).numpy())
        # per-tensor magnitudes are honoured: scales differ across tensors
        assert len(set(scales)) > 1
