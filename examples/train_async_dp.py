#!/usr/bin/env python3
"""Minimal async-DP training demo: N ranks train one GPT-2 against a shared
parameter tensor; the engine gossips compressed deltas in the background.

Single machine (one rank per GPU, or all on CPU):
    python examples/train_async_dp.py --ranks 2 --steps 50

Each rank runs its own process; rank 0 is the tree root.  There is no
lockstep all-reduce anywhere: staleness is bounded by the compression scale
(printed as `stale`), exactly as in the reference's design — this script is
the reference's example.lua grown into a real training loop.
"""
import argparse
import os
import subprocess
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])


def rank_main(args):
    import torch
    from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
    from sharedtensor_amd.parallel.async_dp import AsyncDPTrainer

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dev = torch.device(f"cuda:{rank % torch.cuda.device_count()}") \
        if torch.cuda.is_available() else torch.device("cpu")
    if dev.type == "cuda":
        torch.cuda.set_device(dev)
    torch.manual_seed(1234)  # identical init everywhere; master's seed wins
    cfg = GPT2Config.small() if dev.type == "cuda" else GPT2Config.tiny()
    model = GPT2(cfg).to(dev)
    trainer = AsyncDPTrainer(
        model, port_base=args.port, rank=rank, world=world,
        lr=args.lr, optimizer=args.opt,
        amp_dtype=torch.bfloat16 if dev.type == "cuda" else None,
        param_dtype=torch.bfloat16 if dev.type == "cuda" else torch.float32,
        codec=args.codec, sync_interval_s=0.05, snapshot_join=True)
    B, T = (8, cfg.block_size) if dev.type == "cuda" else (2, 32)
    gen = torch.Generator().manual_seed(rank)
    data = torch.randint(0, cfg.vocab_size, (B, T + 1), generator=gen).to(dev)
    t0 = time.time()
    for step in range(args.steps):
        loss = trainer.step(data[:, :-1], data[:, 1:])
        if step % 10 == 0 or step == args.steps - 1:
            s = trainer.stats()
            print(f"[rank {rank}] step {step:4d} loss {float(loss):.3f} "
                  f"sync_rounds {s['rounds_sent']} "
                  f"stale {s['staleness_p50']}", flush=True)
    print(f"[rank {rank}] {B * T * args.steps / (time.time() - t0):,.0f} "
          f"tokens/s", flush=True)
    trainer.close()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ranks", type=int, default=2)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--lr", type=float, default=0.01)
    ap.add_argument("--opt", choices=["sgd", "adamw"], default="sgd")
    ap.add_argument("--codec", choices=["1bit", "fp8", "int4"], default="1bit")
    ap.add_argument("--port", type=int, default=22000)
    args = ap.parse_args()
    if "RANK" in os.environ:
        rank_main(args)
        return
    procs = []
    for r in range(args.ranks):
        env = dict(os.environ, RANK=str(r), WORLD_SIZE=str(args.ranks))
        procs.append(subprocess.Popen([sys.executable, __file__]
                                      + sys.argv[1:], env=env))
    rc = 0
    for p in procs:
        p.wait()
        rc = rc or p.returncode
    sys.exit(rc)


if __name__ == "__main__":
    main()
