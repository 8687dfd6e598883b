"""Subtree-preserving reconnect (cfg.preserve_subtree, opt-in).

Chain M(root) -- C -- G with explicit topology.  M crashes and restarts;
C (preserve_subtree=True) rejoins carrying G: G's link to C must NEVER
drop (no reconnect, no re-snapshot at G), every update G makes during the
outage survives exactly once everywhere, and all three replicas converge.
The correction algebra lives in engine.cpp (subtree-preserving
reconciliation: corr = S + R - V_old drains to the children as gossip).
"""
import multiprocessing as mp
import os
import time

import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port, wait_until

N = 8192
SEED = 4.0


def _m_proc(port, ready_q, die_ev):
    h = st.SharedTensor("127.0.0.1", port, torch.full((N,), SEED),
                        expected_children=1, provision_up=False,
                        listen_port=port)
    ready_q.put(h.is_master)
    die_ev.wait(120)
    os._exit(1)  # crash: no CLOSE


def _c_proc(port, q, stop_ev):
    try:
        h = st.SharedTensor(
            "127.0.0.1", port, torch.zeros(N), reconnect=True,
            preserve_subtree=True, snapshot_join=True,
            expected_children=1, provision_up=True,
            explicit_parent=f"127.0.0.1:{port}", listen_port=port + 1,
            join_timeout_s=120)
        q.put(("c_up", None))
        while not stop_ev.is_set():
            time.sleep(0.1)
        time.sleep(3)  # drain
        out = torch.zeros(N)
        h.copy_to_tensor(out)
        q.put(("c_final", (float(out[0]), h.stats()["reconnects"],
                           h.stats()["last_error"])))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("c_final", ("exc: " + repr(e), -1, "")))


def _g_proc(port, q, go_ev, done_ev):
    try:
        h = st.SharedTensor(
            "127.0.0.1", port, torch.zeros(N),
            expected_children=0, provision_up=True,
            explicit_parent=f"127.0.0.1:{port + 1}", listen_port=port + 2,
            join_timeout_s=120)
        out = torch.zeros(N)

        def at(v, tol=1e-2):
            h.copy_to_tensor(out)
            return abs(out[0].item() - v) < tol

        if not wait_until(lambda: at(SEED), timeout=60):
            q.put(("g_final", ("never converged initially", -1, False, 0)))
            return
        q.put(("g_ready", None))
        go_ev.wait(60)
        total = 0.0
        while not done_ev.is_set():
            h.add_from_tensor(torch.full((N,), 1e-3))
            total += 1e-3
            time.sleep(0.005)
        # exactly-once across the preserved subtree: final == seed + total
        target = SEED + total
        ok = wait_until(lambda: at(target, tol=0.02), timeout=40)
        s = h.stats()
        up_alive = s["links"][0]["active"] and not s["links"][0]["dead"]
        q.put(("g_final", (float(out[0]), target if ok else -target,
                           up_alive, s["reconnects"])))
        time.sleep(3)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("g_final", ("exc: " + repr(e), -1, False, 0)))


def test_child_carries_grandchild_through_master_restart():
    port = free_port(span=3)
    ctx = mp.get_context("spawn")
    ready_q = ctx.Queue()
    die_ev = ctx.Event()
    q = ctx.Queue()
    stop_ev = ctx.Event()
    go_ev = ctx.Event()
    done_ev = ctx.Event()

    m = ctx.Process(target=_m_proc, args=(port, ready_q, die_ev))
    m.start()
    assert ready_q.get(timeout=30) is True
    c = ctx.Process(target=_c_proc, args=(port, q, stop_ev))
    c.start()
    g = ctx.Process(target=_g_proc, args=(port, q, go_ev, done_ev))
    g.start()
    m2 = None
    try:
        msgs = {}
        for _ in range(2):
            k, v = q.get(timeout=90)
            msgs[k] = v
        assert "c_up" in msgs and "g_ready" in msgs
        time.sleep(1.5)  # drain everything to M before the crash
        die_ev.set()
        m.join(timeout=20)
        go_ev.set()  # G adds through the outage
        time.sleep(1.5)
        die_ev2 = ctx.Event()  # keep a reference: inline Events get GC'd
        m2 = ctx.Process(target=_m_proc, args=(port, ready_q, die_ev2))
        m2.start()
        assert ready_q.get(timeout=60) is True
        time.sleep(6.0)  # C rejoins with G attached; corr drains
        done_ev.set()
        k, (g_val, g_target, up_alive, g_reconnects) = q.get(timeout=90)
        assert k == "g_final"
        assert g_target > 0, f"G did not reach seed+adds: {g_val} vs {-g_target}"
        assert up_alive, "G's link to C dropped — subtree was not preserved"
        assert g_reconnects == 0, "G reconnected; subtree was not preserved"
        stop_ev.set()
        k, (c_val, c_reconnects, c_err) = q.get(timeout=90)
        assert k == "c_final"
        assert isinstance(c_val, float), c_val
        assert abs(c_val - g_target) < 0.03, \
            f"C at {c_val}, G target {g_target} (err={c_err})"
        assert c_reconnects >= 1, "C never rejoined?"
    finally:
        stop_ev.set()
        done_ev.set()
        for p in (c, g, m2):
            if p is not None:
                p.join(timeout=30)
                if p.is_alive():
                    p.kill()
        if m.is_alive():
            m.kill()


def test_correction_algebra_is_exact():
    """Pure simulation of the subtree-preserving reconciliation phases:
    after the child drains its residual, its view equals the rejoiner's
    values EXACTLY, for any V_old, S, R, U_c and any concurrent update u
    landing during the lock-free snapshot window."""
    torch.manual_seed(0)
    n = 1000
    V_old = torch.randn(n)   # rejoiner's pre-reconciliation replica
    S = torch.randn(n)       # new parent's snapshot
    R = torch.randn(n)       # unsent up-residual at capture
    U_c = torch.randn(n)     # child-link residual (still exact vs V_old)
    u = torch.randn(n)       # updates racing the snapshot window
    child_view = V_old - U_c  # drained-invariant starting point

    # phase 1 (exclusive): capture R, corr := -V_old, zero values/up.delta
    tmpR = R.clone()
    corr = -V_old.clone()
    values = torch.zeros(n)
    up_delta = torch.zeros(n)
    child_delta = U_c.clone()  # untouched

    # phase 2 (lock-free): snapshot adds into values+corr; u adds into
    # values and every provisioned residual — never into corr
    values += S
    corr += S
    values += u
    up_delta += u
    child_delta += u

    # phase 3 (exclusive): re-add R; child gets the correction
    values += tmpR
    up_delta += tmpR
    corr += tmpR
    child_delta += corr

    final_child = child_view + child_delta  # residual fully drained
    torch.testing.assert_close(final_child, values, rtol=0, atol=1e-5)
    torch.testing.assert_close(values, S + R + u, rtol=0, atol=1e-5)
    # and the upward residual carries exactly R + u
    torch.testing.assert_close(up_delta, R + u, rtol=0, atol=1e-5)
