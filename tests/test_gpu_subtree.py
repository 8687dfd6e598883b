"""Subtree-preserving reconnect with GPU engines (the HIP branch of the
correction-delta reconciliation): 3 processes share one device, chain
M -- C -- G; M crashes and restarts; C rejoins carrying G."""
import multiprocessing as mp
import os
import time

import pytest
import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port, wait_until

pytestmark = pytest.mark.gpu

N = 1 << 20
SEED = 4.0


def _m_proc(port, ready_q, die_ev):
    torch.cuda.set_device(0)
    h = st.SharedTensor("127.0.0.1", port,
                        torch.full((N,), SEED, device="cuda"),
                        expected_children=1, provision_up=False,
                        listen_port=port)
    ready_q.put(h.is_master)
    die_ev.wait(120)
    os._exit(1)


def _c_proc(port, q, stop_ev):
    try:
        torch.cuda.set_device(0)
        h = st.SharedTensor(
            "127.0.0.1", port, torch.zeros(N, device="cuda"), reconnect=True,
            preserve_subtree=True, snapshot_join=True,
            expected_children=1, provision_up=True,
            explicit_parent=f"127.0.0.1:{port}", listen_port=port + 1,
            join_timeout_s=120)
        q.put(("c_up", None))
        while not stop_ev.is_set():
            time.sleep(0.1)
        time.sleep(3)
        out = torch.zeros(N, device="cuda")
        h.copy_to_tensor(out)
        torch.cuda.synchronize()
        q.put(("c_final", (float(out[0].item()), h.stats()["reconnects"],
                           h.stats()["last_error"])))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("c_final", ("exc: " + repr(e), -1, "")))


def _g_proc(port, q, go_ev, done_ev):
    try:
        torch.cuda.set_device(0)
        h = st.SharedTensor(
            "127.0.0.1", port, torch.zeros(N, device="cuda"),
            expected_children=0, provision_up=True,
            explicit_parent=f"127.0.0.1:{port + 1}", listen_port=port + 2,
            join_timeout_s=120)
        out = torch.zeros(N, device="cuda")

        def at(v, tol=1e-2):
            h.copy_to_tensor(out)
            torch.cuda.synchronize()
            return abs(out[0].item() - v) < tol

        if not wait_until(lambda: at(SEED), timeout=90):
            q.put(("g_final", ("never converged", -1, False, 0)))
            return
        q.put(("g_ready", None))
        go_ev.wait(60)
        total = 0.0
        add = torch.full((N,), 1e-3, device="cuda")
        while not done_ev.is_set():
            h.add_from_tensor(add)
            torch.cuda.synchronize()
            total += 1e-3
            time.sleep(0.01)
        target = SEED + total
        ok = wait_until(lambda: at(target, tol=0.02), timeout=60)
        s = h.stats()
        up_alive = s["links"][0]["active"] and not s["links"][0]["dead"]
        q.put(("g_final", (float(out[0].item()), target if ok else -target,
                           up_alive, s["reconnects"])))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("g_final", ("exc: " + repr(e), -1, False, 0)))


def test_gpu_subtree_preserved_through_master_restart():
    torch.cuda.set_device(0)
    port = free_port(span=3)
    ctx = mp.get_context("spawn")
    ready_q = ctx.Queue()
    die_ev = ctx.Event()
    die_ev2 = ctx.Event()
    q = ctx.Queue()
    stop_ev = ctx.Event()
    go_ev = ctx.Event()
    done_ev = ctx.Event()
    m = ctx.Process(target=_m_proc, args=(port, ready_q, die_ev))
    m.start()
    assert ready_q.get(timeout=90) is True
    c = ctx.Process(target=_c_proc, args=(port, q, stop_ev))
    c.start()
    g = ctx.Process(target=_g_proc, args=(port, q, go_ev, done_ev))
    g.start()
    m2 = None
    try:
        msgs = {}
        for _ in range(2):
            k, v = q.get(timeout=150)
            msgs[k] = v
        assert "c_up" in msgs and "g_ready" in msgs
        time.sleep(1.5)
        die_ev.set()
        m.join(timeout=20)
        go_ev.set()
        time.sleep(1.5)
        m2 = ctx.Process(target=_m_proc, args=(port, ready_q, die_ev2))
        m2.start()
        assert ready_q.get(timeout=90) is True
        time.sleep(6.0)
        done_ev.set()
        k, (g_val, g_target, up_alive, g_reconnects) = q.get(timeout=120)
        assert k == "g_final"
        assert g_target > 0, f"G missed seed+adds: {g_val} vs {-g_target}"
        assert up_alive and g_reconnects == 0, "subtree was not preserved"
        stop_ev.set()
        k, (c_val, c_reconnects, c_err) = q.get(timeout=120)
        assert k == "c_final"
        assert isinstance(c_val, float), c_val
        assert abs(c_val - g_target) < 0.03, (c_val, g_target, c_err)
        assert c_reconnects >= 1
    finally:
        stop_ev.set()
        done_ev.set()
        for p in (c, g, m2):
            if p is not None:
                p.join(timeout=40)
                if p.is_alive():
                    p.kill()
        if m.is_alive():
            m.kill()
