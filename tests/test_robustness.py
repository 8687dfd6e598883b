"""Robustness: non-finite inputs and shutdown under load."""
import multiprocessing as mp
import os
import threading
import time

import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port


def test_nan_input_does_not_crash_engine():
    """A NaN in the residual makes the scale reduction yield 0 (idle link)
    instead of shipping NaN scales; the engine keeps running and finite
    updates still flow once the user repairs their values.  (The reference
    would gossip NaN packets forever.)"""
    port = free_port()
    with st.create_or_fetch("127.0.0.1", port, torch.zeros(64)) as h:
        bad = torch.zeros(64)
        bad[3] = float("nan")
        h.add_from_tensor(bad)
        out = torch.zeros(64)
        h.copy_to_tensor(out)
        assert torch.isnan(out[3])  # NaN propagates to the replica (user bug)
        assert h.stats()["last_error"] == ""
        # engine still answers API calls
        h.add_from_tensor(torch.ones(64))
        h.copy_to_tensor(out)
        assert out[0] == 1.0


def _closer_child(port, q):
    try:
        h = st.create_or_fetch("127.0.0.1", port, torch.zeros(4096))
        # hammer adds from a thread while the main thread closes mid-flight
        stop = threading.Event()

        def spam():
            while not stop.is_set():
                h.add_from_tensor(torch.randn(4096))

        t = threading.Thread(target=spam)
        t.start()
        time.sleep(1.0)
        stop.set()
        t.join()
        h.close()
        q.put(("ok", None))
    except Exception as e:  # pragma: no cover
        q.put(("fail", repr(e)))


def test_close_during_active_gossip():
    port = free_port()
    ctx = mp.get_context("spawn")
    master = st.create_or_fetch("127.0.0.1", port, torch.zeros(4096))
    q = ctx.Queue()
    p = ctx.Process(target=_closer_child, args=(port, q))
    p.start()
    try:
        # master also stays busy
        for _ in range(30):
            master.add_from_tensor(torch.randn(4096))
            time.sleep(0.02)
        status, msg = q.get(timeout=60)
        assert status == "ok", msg
        # master survives the child's departure and keeps serving
        master.add_from_tensor(torch.ones(4096))
        out = torch.zeros(4096)
        master.copy_to_tensor(out)
        assert torch.isfinite(out).all()
    finally:
        p.join(timeout=30)
        master.close()
    assert p.exitcode == 0


def test_many_engine_lifecycles_no_leak():
    """Open/close many engines in one process: no fd/thread leaks."""
    port = free_port()
    for i in range(25):
        with st.create_or_fetch("127.0.0.1", port, torch.randn(256)) as h:
            h.add_from_tensor(torch.ones(256))
    # fd count should stay bounded (threads joined, sockets closed)
    nfds = len(os.listdir("/proc/self/fd"))
    assert nfds < 128, nfds
