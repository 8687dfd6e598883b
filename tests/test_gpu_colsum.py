"""colsum_bf16 numerics vs fp32 torch (bias-grad reduction kernel; measured
SLOWER than torch's reduce at GPT-2 shapes — kept as a tested building
block with the negative result recorded in profiles/README.md)."""
import pytest
import torch

from sharedtensor_amd import _core

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("R,C", [(65536, 768), (1000, 3072), (7, 64)])
def test_matches_fp32_sum(R, C):
    torch.cuda.set_device(0)
    torch.manual_seed(R + C)
    x = torch.randn(R, C, device="cuda").to(torch.bfloat16)
    o = torch.zeros(C, dtype=torch.float32, device="cuda")
    _core.colsum_bf16(x.data_ptr(), o.data_ptr(), R, C,
                      torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    torch.testing.assert_close(o, x.float().sum(0), rtol=1e-3, atol=1e-1)
