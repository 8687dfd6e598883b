"""Chaos/endurance at tree scale: kill and restart interior ranks of an
8-rank tree while every rank keeps adding deltas (reconnect=True).

Extends test_reconnect.py to the full-tree failure mode the reference
cannot survive at all (any disconnect exits the process,
/root/reference/src/sharedtensor.c:62,80,99).  Asserts: the tree heals,
every replica converges to a COMMON state after quiescence (the rejoin
reconciliation V := S + unsent-residual makes delivered-to-dead-parent
updates drop consistently on all replicas), reconnects actually happened,
and no rank leaks file descriptors across the heal cycles.
"""
import multiprocessing as mp
import os
import random
import time

import torch

from sharedtensor_amd.engine import SharedFlat
from sharedtensor_amd.parallel.async_dp import tree_children, tree_parent
from sharedtensor_amd.utils import free_port

N = 4096
WORLD = 8


def _fd_count():
    try:
        return len(os.listdir("/proc/self/fd"))
    except OSError:  # pragma: no cover
        return -1


def _chaos_rank(rank, port_base, stop_ev, q, preserve=False):
    try:
        sh = SharedFlat(
            "127.0.0.1", port_base, [N], device="cpu", codec="1bit",
            reconnect=True, snapshot_join=True, use_rccl=False,
            preserve_subtree=preserve,
            expected_children=len(tree_children(rank, WORLD)),
            provision_up=rank > 0,
            explicit_parent=(f"127.0.0.1:{port_base + tree_parent(rank)}"
                             if rank else ""),
            listen_port=port_base + rank, join_timeout_s=120)
        sh._start()
        fd0 = _fd_count()
        delta = torch.full((N,), 1e-3)
        while not stop_ev.is_set():
            sh._add_flat(delta)
            time.sleep(0.05)
        time.sleep(10.0)  # drain: residuals decay geometrically to ~0
        st = sh.stats()
        q.put(("ok", rank, sh.values.clone(), st["reconnects"], fd0,
               _fd_count(), st["last_error"]))
        time.sleep(4.0)  # keep links alive while peers report
        sh.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", rank, repr(e), 0, 0, 0, ""))


import pytest


@pytest.mark.parametrize("preserve", [False, True],
                         ids=["demote", "preserve_subtree"])
def test_tree_chaos_kill_restart_interior_ranks(preserve):
    port_base = free_port(span=WORLD)
    ctx = mp.get_context("spawn")
    stop_ev = ctx.Event()
    q = ctx.Queue()

    def spawn(rank):
        p = ctx.Process(target=_chaos_rank,
                        args=(rank, port_base, stop_ev, q, preserve))
        p.start()
        return p

    procs = {r: spawn(r) for r in range(WORLD)}
    try:
        time.sleep(6.0)  # tree forms, adds flowing
        # deterministic interior kills (children must rejoin through the
        # restarted parent) + one seeded-random non-root kill
        rng = random.Random(1234)
        for victim in (1, 2, rng.choice(range(3, WORLD))):
            procs[victim].kill()
            procs[victim].join(timeout=10)
            time.sleep(1.0)
            procs[victim] = spawn(victim)
            time.sleep(11.0)  # heal window: rejoin + snapshot + drain

        stop_ev.set()
        reports = []
        for _ in range(WORLD):
            rep = q.get(timeout=120)
            assert rep[0] == "ok", f"rank {rep[1]} failed: {rep[2]}"
            reports.append(rep)
    finally:
        stop_ev.set()
        for p in procs.values():
            p.join(timeout=30)
            if p.is_alive():
                p.kill()

    assert len(reports) == WORLD
    # all replicas converged to one common state
    vals = {r: v for _, r, v, *_ in reports}
    ref = vals[0]
    for r, v in vals.items():
        diff = (v - ref).abs().max().item()
        assert diff < 0.08, f"rank {r} diverged from root by {diff}"
    # the kills really exercised reconnection (long-lived children of the
    # killed interior ranks rejoin and count it)
    total_reconnects = sum(rep[3] for rep in reports)
    assert total_reconnects >= 1, [rep[3] for rep in reports]
    # no fd leak across heal cycles (slack: transient accept/walk sockets)
    for _, r, _, _, fd0, fd1, err in reports:
        assert fd1 <= fd0 + 8, f"rank {r} leaked fds: {fd0} -> {fd1} ({err})"
