"""Async-DP trainer wiring on CPU with 2 processes (explicit tree topology —
the same code path the multi-GPU bench uses, minus the GPU)."""
import multiprocessing as mp
import time

import torch

from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
from sharedtensor_amd.parallel.async_dp import (AsyncDPTrainer, tree_children,
                                                tree_parent)
from sharedtensor_amd.utils import free_port




def test_tree_helpers():
    assert tree_parent(1) == 0 and tree_parent(2) == 0
    assert tree_parent(3) == 1 and tree_parent(4) == 1
    assert tree_children(0, 8) == [1, 2]
    assert tree_children(3, 8) == [7]
    assert tree_children(5, 8) == []


def _worker(rank, world, port_base, q):
    try:
        torch.manual_seed(1234)  # identical init on all ranks
        cfg = GPT2Config.tiny()
        model = GPT2(cfg)
        tr = AsyncDPTrainer(model, host="127.0.0.1", port_base=port_base,
                            rank=rank, world=world, lr=0.1, amp_dtype=None,
                            use_rccl=False)
        assert tr.shared.is_master == (rank == 0)
        g = torch.Generator().manual_seed(7)  # same batch on both ranks
        x = torch.randint(0, cfg.vocab_size, (2, 17), generator=g)
        losses = []
        for _ in range(10):
            loss = tr.step(x[:, :-1], x[:, 1:])
            losses.append(float(loss))
            time.sleep(0.02)
        # training progressed and no link errors
        err = tr.stats()["last_error"]
        rounds = tr.stats()["rounds_sent"] + tr.stats()["rounds_recv"]
        ok = losses[-1] < losses[0] and err == "" and rounds > 0
        q.put(("ok" if ok else "fail",
               f"rank{rank} losses={losses[0]:.3f}->{losses[-1]:.3f} "
               f"rounds={rounds} err={err}"))
        time.sleep(1.0)
        tr.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", f"rank{rank}: {e!r}"))


def test_two_rank_async_dp_cpu():
    port_base = free_port(span=2)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port_base, q))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        results = [q.get(timeout=180) for _ in procs]
        for status, msg in results:
            assert status == "ok", msg
    finally:
        for p in procs:
            p.join(timeout=60)
    for p in procs:
        assert p.exitcode == 0


def test_four_rank_async_dp_cpu():
    """Depth-2 explicit tree (rank 0 <- 1,2; rank 1 <- 3): the same topology
    wiring the driver's 8-GPU scaling bench uses, exercised on CPU."""
    port_base = free_port(span=4)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 4, port_base, q))
             for r in range(4)]
    for p in procs:
        p.start()
    try:
        results = [q.get(timeout=240) for _ in procs]
        for status, msg in results:
            assert status == "ok", msg
    finally:
        for p in procs:
            p.join(timeout=60)
    for p in procs:
        assert p.exitcode == 0
