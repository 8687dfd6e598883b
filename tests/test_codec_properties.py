"""Property-based codec tests (hypothesis): the wire invariants must hold
for arbitrary finite fp32 residuals, including denormals and extremes."""
import math

import numpy as np
import torch
from hypothesis import given, settings, strategies as st_

from sharedtensor_amd import _core
from sharedtensor_amd.ops import oracle as oc

finite_floats = st_.floats(min_value=-float(2.0**100), max_value=float(2.0**100),
                           allow_nan=False, allow_infinity=False,
                           allow_subnormal=False, width=32)


def is_pow2(x):
    if x == 0:
        return True
    m, _ = math.frexp(x)
    return m == 0.5


@settings(max_examples=60, deadline=None)
@given(st_.lists(finite_floats, min_size=1, max_size=300),
       st_.sampled_from([0, 1, 2]))
def test_ef_invariant_arbitrary_values(vals, codec):
    d = torch.tensor(vals, dtype=torch.float32)
    scale, payload, new_d = oc.encode(codec, d)
    assert is_pow2(scale), scale
    assert len(payload) == oc.payload_bytes(codec, d.numel())
    sent = oc.decode(codec, payload, scale, d.numel())
    # the residual is exactly the fp32 rounding of (d - sent)
    np.testing.assert_array_equal(new_d.numpy(), (d - sent).numpy())
    assert torch.isfinite(new_d).all()


@settings(max_examples=40, deadline=None)
@given(st_.lists(finite_floats, min_size=1, max_size=200),
       st_.sampled_from([0, 1, 2]))
def test_cpp_cpu_codec_matches_oracle(vals, codec):
    d = torch.tensor(vals, dtype=torch.float32)
    n = d.numel()
    d2 = d.clone()
    scale_o, payload_o, res_o = oc.encode(codec, d)
    buf = torch.zeros(_core.payload_bytes(codec, n), dtype=torch.uint8)
    scale_c, _ = _core.cpu_encode(codec, d2.data_ptr(), n, -1.0, buf.data_ptr())
    assert scale_c == scale_o
    assert buf.numpy().tobytes() == payload_o
    np.testing.assert_array_equal(d2.numpy(), res_o.numpy())


@settings(max_examples=40, deadline=None)
@given(st_.floats(min_value=-500, max_value=500, allow_nan=False, width=32))
def test_e4m3_conversion_matches_torch(x):
    ours = _core.f32_to_e4m3(float(np.float32(min(max(x, -448.0), 448.0))))
    ref = torch.tensor([x], dtype=torch.float32).clamp(-448, 448) \
        .to(torch.float8_e4m3fn).view(torch.uint8).item()
    assert ours == ref, (x, ours, ref)
