"""RCCL data-plane tests on a single GPU.

What one leased GPU can and cannot execute (documented measurement,
profiles/README.md round 2): RCCL rejects a 2-rank communicator whose ranks
share one device ("Duplicate GPU detected" in librccl), and aborting the
half-made duplicate comm can hang the process — so the cross-process
same-device upgrade is NOT forced here.  Instead this file executes:

  * rccl_self_test      — non-blocking comm init + self all-reduce
  * rccl_loopback_payload — a REAL payload through ncclSend/ncclRecv (1-rank
    self send/recv, grouped), with the exact wait-on-comm-then-stream
    ordering the 2-GPU xGMI links use (csrc/rccl_transport.cpp), verified
    byte-for-byte at several message sizes including an engine-sized one
  * upgrade-failure fallback — SHTENS_TEST_RCCL_FAIL injects a deterministic
    rccl_upgrade failure on both sides; after 2 attempts both peers stop
    negotiating and the link must come up on plain TCP and converge
    (the resilience path a cold multi-GPU run depends on)
"""
import multiprocessing as mp
import os
import time

import pytest
import torch

import sharedtensor_amd as st
from sharedtensor_amd import _core
from sharedtensor_amd.utils import free_port, wait_until

pytestmark = pytest.mark.gpu

N = 1 << 20


def test_rccl_self_allreduce():
    torch.cuda.set_device(0)
    _core.rccl_self_test(0)


@pytest.mark.parametrize("nbytes", [4096, 1 << 20, 131072 + 8])
def test_rccl_loopback_payload(nbytes):
    """Real bytes through ncclSend/ncclRecv on the device, verified."""
    torch.cuda.set_device(0)
    _core.rccl_loopback_payload(0, nbytes)


def test_rccl_loopback_engine_message_size():
    """The engine's exact 1-bit message size for a 1M-element tensor."""
    torch.cuda.set_device(0)
    _core.rccl_loopback_payload(0, 8 + (N // 8))


def _fallback_child(port, q):
    try:
        torch.cuda.set_device(0)
        h = st.create_or_fetch("127.0.0.1", port,
                               torch.zeros(N, device="cuda"),
                               join_timeout_s=60)
        target = torch.full((N,), 3.0, device="cuda")
        out = torch.zeros(N, device="cuda")

        def conv():
            h.copy_to_tensor(out)
            torch.cuda.synchronize()
            return torch.allclose(out, target, atol=1e-2)

        if not wait_until(conv, timeout=60):
            q.put(("fail", f"no converge: {out[:4].cpu()} "
                           f"err={h.stats()['last_error']}"))
            return
        s = h.stats()
        up = s["links"][0]
        if up["rccl"]:
            q.put(("fail", f"link should have fallen back to TCP: {s}"))
            return
        h.add_from_tensor(torch.full((N,), 2.0, device="cuda"))
        q.put(("ok", None))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", repr(e)))


def test_rccl_upgrade_failure_falls_back_to_tcp():
    """Both sides want RCCL, the upgrade fails (injected) twice, and the
    join must still complete over TCP with full data-plane convergence."""
    os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"] = "1"  # negotiate for real
    os.environ["SHTENS_TEST_RCCL_FAIL"] = "1"          # ...and fail it
    try:
        port = free_port()
        ctx = mp.get_context("spawn")
        torch.cuda.set_device(0)
        master = st.create_or_fetch("127.0.0.1", port,
                                    torch.full((N,), 3.0, device="cuda"))
        q = ctx.Queue()
        p = ctx.Process(target=_fallback_child, args=(port, q))
        p.start()
        try:
            status, msg = q.get(timeout=150)
            assert status == "ok", msg
            s = master.stats()
            assert "rccl" in (s["last_error"] or "").lower(), s["last_error"]
            assert not any(l["rccl"] for l in s["links"])
            out = torch.zeros(N, device="cuda")
            target = torch.full((N,), 5.0, device="cuda")

            def conv():
                master.copy_to_tensor(out)
                torch.cuda.synchronize()
                return torch.allclose(out, target, atol=1e-2)

            assert wait_until(conv, timeout=60), \
                f"master: {out[:4].cpu()} stats={master.stats()}"
        finally:
            p.join(timeout=90)
            if p.is_alive():
                p.kill()
            master.close()
        assert p.exitcode == 0
    finally:
        del os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"]
        del os.environ["SHTENS_TEST_RCCL_FAIL"]
