"""RCCL sanity on a single GPU: library + non-blocking comm init + self
all-reduce.  (The 2-GPU xGMI link path is exercised by the driver's
multi-GPU scaling run; rccl_wanted correctly declines same-device pairs,
covered in test_gpu_engine.py.)"""
import pytest
import torch

import sharedtensor_amd  # noqa: F401
from sharedtensor_amd import _core

pytestmark = pytest.mark.gpu


def test_rccl_self_allreduce():
    torch.cuda.set_device(0)
    _core.rccl_self_test(0)
