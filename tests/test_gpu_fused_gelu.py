"""Fused bf16 GELU kernels vs torch's tanh-approximate gelu."""
import pytest
import torch

from sharedtensor_amd.ops import fused_gelu

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("shape", [(64, 3072), (7, 2, 64), (1, 8)])
def test_fwd_bwd_matches_torch(shape):
    torch.manual_seed(sum(shape))
    x = (torch.randn(*shape, device="cuda") * 3).to(torch.bfloat16)
    dy = torch.randn(*shape, device="cuda").to(torch.bfloat16)

    xt = x.clone().requires_grad_(True)
    yt = torch.nn.functional.gelu(xt, approximate="tanh")
    yt.backward(dy)

    xf = x.clone().requires_grad_(True)
    yf = fused_gelu.fused_gelu(xf)
    yf.backward(dy)
    torch.cuda.synchronize()

    torch.testing.assert_close(yf.float(), yt.float(), rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(xf.grad.float(), xt.grad.float(), rtol=5e-2,
                               atol=2e-2)
