"""Engine integration on a single GPU: two processes share cuda:0 and sync
over loopback TCP (the RCCL upgrade correctly declines same-device pairs).
Exercises the full GPU data path: HIP reduce/quantize on the send stream,
pinned staging, H2D + fused unpack/scatter on the recv stream.
"""
import multiprocessing as mp
import time

import pytest
import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port, wait_until

pytestmark = pytest.mark.gpu






def _gpu_child(port, q, codec, lagged=False, bf16d=False):
    try:
        torch.cuda.set_device(0)
        seed = torch.zeros(1 << 20, device="cuda")
        h = st.create_or_fetch("127.0.0.1", port, seed, codec=codec,
                               lagged_scale=lagged,
                               delta_dtype=torch.bfloat16 if bf16d else torch.float32)
        target = torch.full((1 << 20,), 3.0, device="cuda")
        out = torch.zeros_like(seed)

        def conv():
            h.copy_to_tensor(out)
            torch.cuda.synchronize()
            return torch.allclose(out, target, atol=1e-2)

        if not wait_until(conv, timeout=60):
            q.put(("fail", f"no converge: {out[:4].cpu()} err={h.stats()['last_error']}"))
            return
        h.add_from_tensor(torch.full_like(seed, 2.0))
        q.put(("ok", None))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", repr(e)))


@pytest.mark.parametrize("codec,lagged,bf16d",
                         [("1bit", False, False), ("fp8", False, False),
                          ("1bit", True, False), ("int4", True, False),
                          ("1bit", True, True), ("fp8", False, True)])
def test_two_process_one_gpu_tcp(codec, lagged, bf16d):
    port = free_port()
    ctx = mp.get_context("spawn")
    torch.cuda.set_device(0)
    master = st.create_or_fetch("127.0.0.1", port,
                                torch.full((1 << 20,), 3.0, device="cuda"),
                                codec=codec, lagged_scale=lagged,
                                delta_dtype=torch.bfloat16 if bf16d else torch.float32)
    q = ctx.Queue()
    p = ctx.Process(target=_gpu_child, args=(port, q, codec, lagged, bf16d))
    p.start()
    try:
        status, msg = q.get(timeout=120)
        assert status == "ok", msg
        out = torch.zeros(1 << 20, device="cuda")
        target = torch.full((1 << 20,), 5.0, device="cuda")

        def conv():
            master.copy_to_tensor(out)
            torch.cuda.synchronize()
            return torch.allclose(out, target, atol=1e-2)

        assert wait_until(conv, timeout=60), \
            f"master: {out[:4].cpu()} stats={master.stats()}"
        # links must NOT have upgraded to RCCL (same device)
        assert not any(l["rccl"] for l in master.stats()["links"])
    finally:
        p.join(timeout=60)
        master.close()
    assert p.exitcode == 0


def test_gpu_master_only_and_fused_sgd():
    torch.cuda.set_device(0)
    port = free_port()
    seed = torch.randn(1 << 16, device="cuda")
    with st.create_or_fetch("127.0.0.1", port, seed.clone()) as h:
        mom = torch.zeros(1 << 16, device="cuda")
        grad = torch.randn(1 << 16, device="cuda")
        h.fused_sgd_step(mom, grad, lr=0.5, momentum=0.0)
        out = torch.zeros_like(seed)
        h.copy_to_tensor(out)
        torch.cuda.synchronize()
        torch.testing.assert_close(out, seed - 0.5 * grad)


def test_gpu_table_from_module():
    torch.cuda.set_device(0)
    port = free_port()
    lin = torch.nn.Linear(64, 32).cuda()
    with st.SharedTable.from_module("127.0.0.1", port, lin) as h:
        v = h.views()
        torch.cuda.synchronize()
        torch.testing.assert_close(v["weight"], lin.weight.data)
