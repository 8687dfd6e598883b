"""Join-storm: concurrent joiners against a master whose snapshot stream is
artificially slowed (SHTENS_TEST_SNAPSHOT_DELAY_MS, test-only knob).

Round-1 weakness #4: `accept_child` used to stream the full snapshot inline
on the listen thread while holding the slot mutex, so a multi-GB snapshot
serialized every other join and redirect.  Joins now run on per-slot
handshake threads; this test proves two snapshot windows overlap in time.
"""
import multiprocessing as mp
import os
import time

import torch

import sharedtensor_amd as st
from sharedtensor_amd.utils import free_port, wait_until

N = 120_000  # 480 KB snapshot; 16 KB test chunks + delay => ~1.4 s/stream


def _joiner(port, q, idx):
    try:
        seed = torch.zeros(N, dtype=torch.float32)
        t0 = time.monotonic()
        h = st.SharedTensor("127.0.0.1", port, seed, snapshot_join=True)
        out = torch.zeros(N)

        def converged():
            h.copy_to_tensor(out)
            return abs(out[0].item() - 3.0) < 1e-3 and \
                abs(out[-1].item() - 7.0) < 1e-3
        ok = wait_until(converged, timeout=90)
        t1 = time.monotonic()
        if not ok:
            q.put(("fail", idx, f"never converged: {out[:3]}", 0.0, 0.0))
            return
        q.put(("ok", idx, None, t0, t1))
        time.sleep(2)  # stay joinable for later walkers
        h.close()
    except Exception as e:  # pragma: no cover
        q.put(("fail", idx, repr(e), 0.0, 0.0))


def test_join_storm_snapshots_overlap():
    os.environ["SHTENS_TEST_SNAPSHOT_DELAY_MS"] = "45"
    try:
        port = free_port()
        seed = torch.zeros(N, dtype=torch.float32)
        seed[0] = 3.0
        seed[-1] = 7.0
        master = st.SharedTensor("127.0.0.1", port, seed, snapshot_join=True)
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_joiner, args=(port, q, i))
                 for i in range(4)]
        for p in procs:
            p.start()
        results = []
        try:
            for _ in procs:
                status, idx, msg, t0, t1 = q.get(timeout=120)
                assert status == "ok", f"joiner {idx}: {msg}"
                results.append((t0, t1))
        finally:
            for p in procs:
                p.join(timeout=30)
                if p.is_alive():
                    p.kill()
            master.close()
        # proof of concurrency: some pair of join windows overlaps by a
        # large fraction of the shorter window (impossible when snapshots
        # were serialized through the listen thread)
        best = 0.0
        for i in range(len(results)):
            for j in range(i + 1, len(results)):
                (a0, a1), (b0, b1) = results[i], results[j]
                ov = min(a1, b1) - max(a0, b0)
                shorter = min(a1 - a0, b1 - b0)
                if shorter > 0:
                    best = max(best, ov / shorter)
        assert best > 0.5, f"join windows never overlapped (best={best:.2f})"
    finally:
        del os.environ["SHTENS_TEST_SNAPSHOT_DELAY_MS"]
