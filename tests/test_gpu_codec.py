"""CDNA4 codec kernel numerics vs the fp32 oracle (plain PyTorch/numpy CPU).

Every kernel output is compared bitwise (payload bytes, residual floats)
against ops/oracle.py on identical inputs.
"""
import numpy as np
import pytest
import torch

import sharedtensor_amd  # noqa: F401
from sharedtensor_amd import _core
from sharedtensor_amd.ops import oracle as oc

pytestmark = pytest.mark.gpu


def stream():
    return torch.cuda.current_stream().cuda_stream


@pytest.mark.parametrize("codec", [0, 1, 2])
@pytest.mark.parametrize("n", [64, 1000, 1 << 20, (1 << 22) + 17])
def test_encode_matches_oracle(codec, n):
    torch.manual_seed(n * 10 + codec)
    d = (torch.randn(n) * (10.0 ** torch.randint(-2, 3, (n,)).float())).cuda()
    d_cpu = d.cpu()
    dc = _core.DevCodec(codec, [n], 0)
    scales = torch.zeros(1, dtype=torch.float32, device="cuda")
    payload = torch.zeros(_core.payload_bytes(codec, n), dtype=torch.uint8,
                          device="cuda")
    dc.reduce_scales(d.data_ptr(), scales.data_ptr(), 1, stream())
    dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), stream())
    torch.cuda.synchronize()
    scale_o, payload_o, res_o = oc.encode(codec, d_cpu)
    assert scales.item() == scale_o
    assert payload.cpu().numpy().tobytes() == payload_o
    np.testing.assert_array_equal(d.cpu().numpy(), res_o.numpy())


@pytest.mark.parametrize("codec", [0, 1, 2])
@pytest.mark.parametrize("ndst", [1, 2, 3])
def test_apply_matches_oracle(codec, ndst):
    n = 1 << 18
    torch.manual_seed(codec * 7 + ndst)
    d = (torch.randn(n) * 2).cuda()
    dc = _core.DevCodec(codec, [n], 0)
    scales = torch.zeros(1, dtype=torch.float32, device="cuda")
    payload = torch.zeros(_core.payload_bytes(codec, n), dtype=torch.uint8,
                          device="cuda")
    dc.reduce_scales(d.data_ptr(), scales.data_ptr(), 1, stream())
    dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), stream())
    dsts = [torch.zeros(n, device="cuda") for _ in range(ndst)]
    dc.apply(payload.data_ptr(), scales.data_ptr(),
             [t.data_ptr() for t in dsts], stream())
    torch.cuda.synchronize()
    dec_o = oc.decode(codec, payload.cpu().numpy().tobytes(), scales.item(), n)
    for t in dsts:
        np.testing.assert_array_equal(t.cpu().numpy(), dec_o.numpy())


def test_table_mode_matches_oracle():
    sizes = [100, 64, 4096, 1]
    torch.manual_seed(5)
    parts = [torch.randn(s) * (10.0 ** i) for i, s in enumerate(sizes)]
    flat = torch.cat(parts).cuda()
    dc = _core.DevCodec(0, sizes, 0)
    T = len(sizes)
    scales = torch.zeros(T, dtype=torch.float32, device="cuda")
    pb = sum(_core.payload_bytes(0, s) for s in sizes)
    payload = torch.zeros(pb, dtype=torch.uint8, device="cuda")
    dc.reduce_scales(flat.data_ptr(), scales.data_ptr(), 1, stream())
    dc.quantize(flat.data_ptr(), scales.data_ptr(), payload.data_ptr(), stream())
    torch.cuda.synchronize()
    scales_o, payload_o, res_o = oc.encode_table(0, parts)
    np.testing.assert_array_equal(scales.cpu().numpy(),
                                  np.array(scales_o, dtype=np.float32))
    assert payload.cpu().numpy().tobytes() == payload_o
    np.testing.assert_array_equal(flat.cpu().numpy(),
                                  torch.cat(res_o).numpy())


def test_add_scatter_and_fused_sgd():
    n = 1 << 20
    torch.manual_seed(9)
    src = torch.randn(n, device="cuda")
    d0, d1 = torch.randn(n, device="cuda"), torch.randn(n, device="cuda")
    d0c, d1c = d0.clone(), d1.clone()
    _core.gpu_add_scatter(src.data_ptr(), n, 2.5,
                          [d0.data_ptr(), d1.data_ptr()], stream())
    torch.cuda.synchronize()
    torch.testing.assert_close(d0, d0c + 2.5 * src, rtol=0, atol=0)
    torch.testing.assert_close(d1, d1c + 2.5 * src, rtol=0, atol=0)

    mom = torch.randn(n, device="cuda")
    grad = torch.randn(n, device="cuda")
    vals = torch.randn(n, device="cuda")
    delta = torch.zeros(n, device="cuda")
    mom_ref = mom.clone()
    vals_ref = vals.clone()
    lr, mu = 0.1, 0.9
    _core.gpu_fused_sgd(mom.data_ptr(), grad.data_ptr(), lr, mu, n,
                        [vals.data_ptr(), delta.data_ptr()], stream())
    torch.cuda.synchronize()
    # the kernel computes m = fma(mu, mom, grad) with a single rounding; the
    # two-rounding torch reference differs by <= 1 ulp
    m_new = mu * mom_ref + grad
    u = -lr * m_new
    torch.testing.assert_close(mom, m_new, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(vals, vals_ref + u, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(delta, u, rtol=1e-5, atol=1e-6)


def test_fused_sgd_bf16():
    n = 1 << 18
    torch.manual_seed(11)
    mom = torch.randn(n, device="cuda")
    grad = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    vals = torch.randn(n, device="cuda")
    delta = torch.zeros(n, device="cuda")
    shadow = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    mom_ref, vals_ref = mom.clone(), vals.clone()
    lr, mu = 0.1, 0.9
    # exercise via DevCodec-less direct binding through an engine-free call:
    from sharedtensor_amd import _core
    # use the plain binding via a tiny Engine is heavyweight; the kernel is
    # covered through bench/smoke; here check semantics with the raw launcher
    # exposed on Engine — fall back to building a master-only shared tensor.
    import sharedtensor_amd as st
    import socket
    s = socket.socket(); s.bind(("127.0.0.1", 0)); port = s.getsockname()[1]; s.close()
    h = st.create_or_fetch("127.0.0.1", port, vals_ref.clone())
    h.fused_sgd_bf16_step(mom, grad, shadow, lr, mu)
    out = torch.zeros(n, device="cuda")
    h.copy_to_tensor(out)
    torch.cuda.synchronize()
    m_new = mu * mom_ref + grad.float()
    u = -lr * m_new
    torch.testing.assert_close(mom, m_new, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(out, vals_ref + u, rtol=1e-5, atol=1e-6)
    # the kernel's fp32 (old + u) can differ from torch's by 1 fp32 ulp,
    # which at a bf16 rounding boundary becomes 1 bf16 ulp
    torch.testing.assert_close(shadow.float(), (vals_ref + u).to(torch.bfloat16).float(),
                               rtol=8e-3, atol=1e-6)
    h.close()


@pytest.mark.parametrize("codec", [0, 1, 2])
def test_lagged_scale_stats(codec):
    """quantize(stats=...) + finalize must equal a fresh reduce over the
    post-quantize residual."""
    n = 1 << 18
    torch.manual_seed(21 + codec)
    d = (torch.randn(n) * 2).cuda()
    dc = _core.DevCodec(codec, [n], 0)
    scales = torch.zeros(1, dtype=torch.float32, device="cuda")
    stats = torch.zeros(1, dtype=torch.float64, device="cuda")  # 8 bytes
    payload = torch.zeros(_core.payload_bytes(codec, n), dtype=torch.uint8,
                          device="cuda")
    s = stream()
    dc.reduce_scales(d.data_ptr(), scales.data_ptr(), 1, s)
    dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), s,
                stats.data_ptr())
    lagged = torch.zeros(1, dtype=torch.float32, device="cuda")
    dc.finalize_scales(stats.data_ptr(), lagged.data_ptr(), s)
    # fresh exact reduce over the updated residual
    fresh = torch.zeros(1, dtype=torch.float32, device="cuda")
    dc.reduce_scales(d.data_ptr(), fresh.data_ptr(), 1, s)
    torch.cuda.synchronize()
    assert lagged.item() == fresh.item(), (lagged.item(), fresh.item())


@pytest.mark.parametrize("codec", [0, 1, 2])
def test_bf16_delta_roundtrip(codec):
    """bf16 residual storage: the EF debit quanta are exactly representable
    in bf16 (pow2 scales x small-mantissa q), so residual' == bf16(residual)
    - sent exactly when the pre-quantize residual is itself bf16."""
    n = 1 << 16
    torch.manual_seed(31 + codec)
    d32 = (torch.randn(n) * 2)
    d = d32.to(torch.bfloat16).cuda()
    d_orig = d.cpu().clone()  # quantize debits d in place
    dc = _core.DevCodec(codec, [n], 0, delta_bf16=True)
    scales = torch.zeros(1, dtype=torch.float32, device="cuda")
    payload = torch.zeros(_core.payload_bytes(codec, n), dtype=torch.uint8,
                          device="cuda")
    dc.reduce_scales(d.data_ptr(), scales.data_ptr(), 1, stream())
    dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), stream())
    torch.cuda.synchronize()
    s = scales.item()
    assert s > 0
    dec = oc.decode(codec, payload.cpu().numpy().tobytes(), s, n)
    # conservation in bf16: residual_after == bf16(fl32(bf16(d)) - sent)
    expect = (d_orig.float() - dec).to(torch.bfloat16)
    assert torch.equal(d.cpu(), expect), \
        (d.cpu().float() - expect.float()).abs().max()
    # and apply into a bf16 destination accumulates
    dst = torch.zeros(n, dtype=torch.bfloat16, device="cuda")
    vals = torch.zeros(n, dtype=torch.float32, device="cuda")
    dc.apply(payload.data_ptr(), scales.data_ptr(),
             [vals.data_ptr(), dst.data_ptr()], stream())
    torch.cuda.synchronize()
    torch.testing.assert_close(vals.cpu(), dec, rtol=0, atol=0)
    torch.testing.assert_close(dst.cpu().float(), dec.to(torch.bfloat16).float(),
                               rtol=0, atol=0)


def test_quantize_keepalive_zero_scale():
    n = 4096
    d = torch.zeros(n, device="cuda")
    dc = _core.DevCodec(0, [n], 0)
    scales = torch.full((1,), 7.0, dtype=torch.float32, device="cuda")
    payload = torch.full((_core.payload_bytes(0, n),), 0xFF, dtype=torch.uint8,
                         device="cuda")
    dc.reduce_scales(d.data_ptr(), scales.data_ptr(), 1, stream())
    dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), stream())
    torch.cuda.synchronize()
    assert scales.item() == 0.0
    assert torch.all(payload == 0)  # zero-filled packet
    assert torch.all(d == 0)
