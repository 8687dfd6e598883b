#!/usr/bin/env python3
"""GPU-engine tree chaos rehearsal: 8 ranks (all GPU engines sharing one
device), kill + restart interior ranks under continuous fused-SGD-style
adds, then assert every replica converges to one common state.

The CPU twin runs in CI (tests/test_chaos.py); this script exercises the
same failure paths with HIP resources in play (link streams, device
buffers, reclaim/rebuild kernels) and reports a JSON verdict for profiles/.
"""
import json
import multiprocessing as mp
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

N = 1 << 20
WORLD = 8


def rank_main(rank, port_base, stop_ev, q):
    from sharedtensor_amd.engine import SharedFlat
    from sharedtensor_amd.parallel.async_dp import tree_children, tree_parent
    try:
        torch.cuda.set_device(0)
        sh = SharedFlat(
            "127.0.0.1", port_base, [N], device="cuda:0", codec="1bit",
            reconnect=True, snapshot_join=True,
            expected_children=len(tree_children(rank, WORLD)),
            provision_up=rank > 0,
            explicit_parent=(f"127.0.0.1:{port_base + tree_parent(rank)}"
                             if rank else ""),
            listen_port=port_base + rank, join_timeout_s=180,
            sync_interval_s=0.02)
        sh._start()
        delta = torch.full((N,), 1e-3, device="cuda")
        while not stop_ev.is_set():
            sh._add_flat(delta)
            torch.cuda.synchronize()
            time.sleep(0.05)
        time.sleep(8.0)  # drain
        st = sh.stats()
        vals = sh.values[:8].cpu().tolist() + [float(sh.values.sum().cpu())]
        q.put(("ok", rank, vals, st["reconnects"], st["last_error"]))
        time.sleep(4.0)
        sh.close()
    except Exception as e:
        q.put(("fail", rank, repr(e), 0, ""))


def main():
    from sharedtensor_amd.utils import free_port
    port_base = free_port()
    ctx = mp.get_context("spawn")
    stop_ev = ctx.Event()
    q = ctx.Queue()

    def spawn(r):
        p = ctx.Process(target=rank_main, args=(r, port_base, stop_ev, q))
        p.start()
        return p

    procs = {r: spawn(r) for r in range(WORLD)}
    time.sleep(15.0)  # tree forms (8 CUDA contexts on one device)
    t0 = time.time()
    for victim in (1, 2):
        procs[victim].kill()
        procs[victim].join(timeout=10)
        time.sleep(1.0)
        procs[victim] = spawn(victim)
        time.sleep(12.0)  # heal: rejoin + snapshot + drain
    stop_ev.set()
    reports = []
    for _ in range(WORLD):
        reports.append(q.get(timeout=180))
    for p in procs.values():
        p.join(timeout=40)
        if p.is_alive():
            p.kill()
    fails = [r for r in reports if r[0] != "ok"]
    sums = sorted(r[2][-1] for r in reports if r[0] == "ok")
    spread = (sums[-1] - sums[0]) if sums else float("nan")
    out = {
        "metric": "gpu-engine 8-rank chaos rehearsal (kill+restart ranks 1,2)",
        "world": WORLD,
        "numel": N,
        "fails": [r[1] for r in fails],
        "replica_sum_spread": round(spread, 3),
        "per_elem_spread": spread / N if sums else None,
        "reconnects_total": sum(r[3] for r in reports if r[0] == "ok"),
        "wall_s": round(time.time() - t0, 1),
        "converged": bool(not fails and spread / N < 0.05),
    }
    print(json.dumps(out))
    return 0 if out["converged"] else 1


if __name__ == "__main__":
    sys.exit(main())
