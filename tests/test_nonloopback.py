"""Tree join + convergence over the host's REAL (non-loopback) interface.

Everything else in CI runs on 127.0.0.1; inter-node deployment talks over
real NICs, where the reference's self-addressing trick (bind the listener
to the local address of the up socket, sharedtensor.c:292-316) sees a
routable address instead of loopback.  Skipped when the environment has no
non-loopback IPv4."""
import multiprocessing as mp
import socket
import time

import pytest
import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port, wait_until


def _primary_ip():
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("10.255.255.255", 1))
        ip = s.getsockname()[0]
    except OSError:
        ip = None
    finally:
        s.close()
    return None if ip in (None, "127.0.0.1") else ip


def _child(host, port, q):
    try:
        h = st.create_or_fetch(host, port, torch.zeros(4096))
        out = torch.zeros(4096)

        def conv():
            h.copy_to_tensor(out)
            return abs(out[0].item() - 11.0) < 1e-2
        ok = wait_until(conv, timeout=45)
        if not ok:
            q.put(("fail", f"no converge: {out[:3]} "
                           f"err={h.stats()['last_error']}"))
            return
        # the walk must hand out the REAL address in redirects: our listen
        # address must be on the same interface, not 127.0.0.1
        q.put(("ok", h.stats()["listen_port"]))
        h.add_from_tensor(torch.full((4096,), 2.0))
        time.sleep(2)
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", repr(e)))


def _walker(host, port, q, idx):
    try:
        h = st.create_or_fetch(host, port, torch.zeros(4096),
                               join_timeout_s=60)
        out = torch.zeros(4096)

        def conv():
            h.copy_to_tensor(out)
            return abs(out[0].item() - 11.0) < 1e-2
        q.put(("ok", idx) if wait_until(conv, timeout=60)
              else ("fail", f"{idx}: no converge"))
        time.sleep(3)  # stay up while later walkers redirect through us
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", f"{idx}: {e!r}"))


def test_redirect_walk_carries_real_addresses():
    """With the master's two slots taken, the third joiner is REDIRECTED:
    the 'N' reply carries a child's observed peer address, which on a real
    interface must be its routable (ip, listen port) — the self-addressing
    trick end-to-end off loopback (sharedtensor.c:224-234,292-316)."""
    ip = _primary_ip()
    if ip is None:
        pytest.skip("no non-loopback IPv4 available")
    port = free_port()
    seed = torch.zeros(4096)
    seed[0] = 11.0
    master = st.SharedTensor(ip, port, seed)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_walker, args=(ip, port, q, i))
             for i in range(3)]
    for p in procs:
        p.start()
    try:
        for _ in procs:
            status, info = q.get(timeout=90)
            assert status == "ok", info
        # 2 direct children + 1 redirected grandchild
        active = [l for l in master.stats()["links"] if l["active"]]
        assert len(active) == 2, master.stats()
        for l in active:
            assert not l["peer"].startswith("127."), l["peer"]
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.kill()
        master.close()


def test_join_and_converge_on_real_interface():
    ip = _primary_ip()
    if ip is None:
        pytest.skip("no non-loopback IPv4 available")
    port = free_port()
    seed = torch.zeros(4096)
    seed[0] = 11.0
    master = st.SharedTensor(ip, port, seed)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_child, args=(ip, port, q))
    p.start()
    try:
        status, info = q.get(timeout=60)
        assert status == "ok", info
        out = torch.zeros(4096)

        def conv():
            master.copy_to_tensor(out)
            return abs(out[0].item() - 13.0) < 1e-2

        assert wait_until(conv, timeout=45), \
            f"master missed child delta: {out[:3]} {master.stats()}"
    finally:
        p.join(timeout=30)
        if p.is_alive():
            p.kill()
        master.close()
    assert p.exitcode == 0
