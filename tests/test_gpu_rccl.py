"""RCCL data-plane tests on a single GPU.

test_rccl_self_allreduce: library + non-blocking comm init + self all-reduce.

The forced same-device tests drive the REAL ncclSend/ncclRecv payload path
(wait_enqueued ordering, stream polling, abort during in-flight p2p,
teardown) between two processes sharing one leased GPU via the test-only
SHTENS_RCCL_FORCE_SAME_DEVICE override — the semantics the 2-GPU xGMI links
use verbatim (engine.cpp rccl_wanted; reference transport semantics
sharedtensor.c:113-131,145-179)."""
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


def _rccl_child(port, q, codec, crash_master_ev=None):
    try:
        torch.cuda.set_device(0)
        h = st.create_or_fetch("127.0.0.1", port,
                               torch.zeros(N, device="cuda"), codec=codec)
        target = torch.full((N,), 3.0, device="cuda")
        out = torch.zeros(N, device="cuda")

        def conv():
            h.copy_to_tensor(out)
            torch.cuda.synchronize()
            return torch.allclose(out, target, atol=1e-2)

        if not wait_until(conv, timeout=90):
            q.put(("fail", f"no converge: {out[:4].cpu()} "
                           f"err={h.stats()['last_error']}"))
            return
        up = h.stats()["links"][0]
        if not up["rccl"]:
            q.put(("fail", f"up link did not upgrade to RCCL: {h.stats()}"))
            return
        h.add_from_tensor(torch.full((N,), 2.0, device="cuda"))
        q.put(("ok", None))
        if crash_master_ev is not None:
            crash_master_ev.wait(60)
            # master just died without CLOSE: the ctrl thread must detect
            # it and abort the in-flight RCCL recv instead of hanging
            ok = wait_until(lambda: h.stats()["links"][0]["dead"], timeout=60)
            q.put(("dead_detected", ok) if ok else
                  ("fail", f"rccl link death undetected: {h.stats()}"))
        time.sleep(2)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", repr(e)))


@pytest.mark.parametrize("codec", ["1bit", "int4"])
def test_forced_same_device_rccl_payload(codec):
    """Payload moves over ncclSend/ncclRecv end-to-end: child receives the
    master's state and the master receives the child's delta, both sides on
    the RCCL data plane."""
    os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"] = "1"
    try:
        port = free_port()
        ctx = mp.get_context("spawn")
        torch.cuda.set_device(0)
        master = st.create_or_fetch("127.0.0.1", port,
                                    torch.full((N,), 3.0, device="cuda"),
                                    codec=codec)
        q = ctx.Queue()
        p = ctx.Process(target=_rccl_child, args=(port, q, codec))
        p.start()
        try:
            status, msg = q.get(timeout=150)
            assert status == "ok", msg
            links = master.stats()["links"]
            assert any(l["rccl"] and l["active"] for l in links), links
            out = torch.zeros(N, device="cuda")
            target = torch.full((N,), 5.0, device="cuda")

            def conv():
                master.copy_to_tensor(out)
                torch.cuda.synchronize()
                return torch.allclose(out, target, atol=1e-2)

            assert wait_until(conv, timeout=90), \
                f"master: {out[:4].cpu()} stats={master.stats()}"
            s = master.stats()
            assert s["rounds_recv"] > 0 and s["bytes_recv"] > 0
        finally:
            p.join(timeout=90)
            master.close()
        assert p.exitcode == 0
    finally:
        del os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"]


def _rccl_mortal_master(port, ready_q, die_ev):
    torch.cuda.set_device(0)
    h = st.create_or_fetch("127.0.0.1", port,
                           torch.full((N,), 3.0, device="cuda"))
    ready_q.put(h.is_master)
    die_ev.wait(120)
    os._exit(1)  # crash: no CLOSE packet, RCCL peer vanishes mid-flight


def test_forced_same_device_rccl_abort_on_peer_death():
    """A dead RCCL peer must be detected (TCP ctrl read fails -> link_down
    -> ncclCommAbort) without hanging the recv loop."""
    os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"] = "1"
    try:
        port = free_port()
        ctx = mp.get_context("spawn")
        ready_q = ctx.Queue()
        die_ev = ctx.Event()
        m = ctx.Process(target=_rccl_mortal_master,
                        args=(port, ready_q, die_ev))
        m.start()
        assert ready_q.get(timeout=90) is True
        q = ctx.Queue()
        crash_ev = ctx.Event()
        c = ctx.Process(target=_rccl_child, args=(port, q, "1bit", crash_ev))
        c.start()
        try:
            status, msg = q.get(timeout=150)
            assert status == "ok", msg
            die_ev.set()
            m.join(timeout=30)
            crash_ev.set()
            status, ok = q.get(timeout=120)
            assert status == "dead_detected" and ok, (status, ok)
        finally:
            c.join(timeout=90)
            if c.is_alive():
                c.kill()
            if m.is_alive():
                m.kill()
        assert c.exitcode == 0
    finally:
        del os.environ["SHTENS_RCCL_FORCE_SAME_DEVICE"]
