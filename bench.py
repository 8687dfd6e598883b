#!/usr/bin/env python3
"""Flagship benchmark: async-DP GPT-2-small training over the shared-tensor
engine (BASELINE.json config 3), plus a param-sync bandwidth mode.

Contract (driver): `python bench.py --gpus N --steps K --warmup W` runs one
rank per GPU.  Two launch paths, both supported:
  * torchrun (`torch.distributed.run --nproc-per-node N`): each rank reads
    RANK / WORLD_SIZE / LOCAL_RANK from the env.
  * plain `python bench.py --gpus N` with no RANK in the env: this process
    SELF-LAUNCHES N rank subprocesses (env RANK/LOCAL_RANK/WORLD_SIZE/
    MASTER_*), waits for them, and forwards rank 0's JSON line.
Each rank does W untimed warmup steps, times exactly K steps bracketed by
barrier + torch.cuda.synchronize on both sides, takes the MAX step time over
ranks, and rank 0 prints ONE JSON line.

The metric is the whole-job aggregate tokens/s; config reports the engine's
param-sync wire GB/s and p50 staleness (the per-round scale: every packet
moves each element by exactly +-scale).  torch.distributed (gloo) is used
ONLY for barriers and the max-reduce of the measured time — the parameter
sync data plane is this framework's engine (TCP tree / RCCL over xGMI).
"""
import argparse
import json
import os
import sys
import time

# multi-process GPU tensor sharing / RCCL need dmabuf IPC on this pool
os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

# the contract is ONE JSON line on stdout from rank 0; gloo/torch print
# banners to fd 1, so reroute fd 1 -> stderr and keep the real stdout for
# the final JSON write
_REAL_STDOUT = os.dup(1)
os.dup2(2, 1)

import torch  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def emit(result: dict):
    os.write(_REAL_STDOUT, (json.dumps(result) + "\n").encode())


def log(msg):
    r = os.environ.get("RANK", "0")
    print(f"[bench r{r}] {msg}", file=sys.stderr, flush=True)


def self_launch(args) -> int:
    """Spawn one rank subprocess per GPU and forward rank 0's JSON line.

    Used when the driver invokes `python bench.py --gpus N` as a single
    command (no torchrun): the rendezvous env torchrun would provide is
    synthesized here.  Returns the exit code for the parent process.
    """
    import signal
    import socket
    import subprocess

    with socket.socket() as s:  # free rendezvous port for the gloo group
        s.bind(("127.0.0.1", 0))
        master_port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env.update({"MASTER_ADDR": "127.0.0.1",
                     "MASTER_PORT": str(master_port),
                     "WORLD_SIZE": str(args.gpus)})
    procs = []
    argv = [sys.executable, os.path.abspath(__file__)] + sys.argv[1:]
    for r in range(args.gpus):
        env = dict(base_env)
        env["RANK"] = str(r)
        env["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen(
            argv, env=env,
            stdout=subprocess.PIPE if r == 0 else sys.stderr,
            start_new_session=True))
    log(f"self-launched {args.gpus} rank processes (master port {master_port})")
    deadline = time.time() + float(os.environ.get("SHTENS_LAUNCH_TIMEOUT", 3600))
    rc = 0
    try:
        for r, p in enumerate(procs):
            remaining = max(1.0, deadline - time.time())
            try:
                p.wait(timeout=remaining)
            except subprocess.TimeoutExpired:
                log(f"rank {r} exceeded launch timeout; killing the job")
                rc = 124
                break
            if p.returncode != 0:
                log(f"rank {r} exited with {p.returncode}")
                rc = rc or p.returncode
    finally:
        for p in procs:
            if p.poll() is None:
                try:  # kill the whole rank's session (it may have children)
                    os.killpg(p.pid, signal.SIGKILL)
                except OSError:
                    p.kill()
    out = procs[0].stdout.read() if procs[0].stdout else b""
    if rc == 0:
        lines = [l for l in out.decode(errors="replace").splitlines() if l.strip()]
        if not lines:
            log("rank 0 produced no JSON line")
            return 1
        os.write(_REAL_STDOUT, (lines[-1] + "\n").encode())
    return rc


def dist_setup(world):
    import torch.distributed as dist
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group("gloo", rank=int(os.environ.get("RANK", 0)),
                                world_size=world)
    return dist if world > 1 else None


def barrier(dist):
    if dist is not None:
        dist.barrier()


def max_over_ranks(dist, x: float) -> float:
    if dist is None:
        return x
    t = torch.tensor([x], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t.item()


def sum_over_ranks(dist, x: float) -> float:
    if dist is None:
        return x
    t = torch.tensor([x], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t.item()


def enable_tunableop(device):
    """Load the checked-in TunableOp GEMM selections (tuned on MI355X for
    the bench shapes; +2.7% measured over the default rocBLAS picks,
    profiles/r02).  Tuning stays OFF — unknown shapes use the defaults;
    a validator mismatch (other GPU/ROCm) makes torch ignore the file."""
    if device.type != "cuda" or os.environ.get("SHTENS_TUNABLEOP", "1") != "1":
        return
    fn = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "sharedtensor_amd", "tunableop_gfx950.csv")
    if not os.path.exists(fn):
        return
    try:
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(fn)
        log("TunableOp: loaded tuned GEMM table")
    except Exception as e:
        log(f"TunableOp unavailable: {e}")


def run_train(args, rank, world, device):
    from sharedtensor_amd.models.gpt2 import GPT2, GPT2Config
    from sharedtensor_amd.models.llama import Llama, LlamaConfig
    from sharedtensor_amd.parallel.async_dp import AsyncDPTrainer

    enable_tunableop(device)
    llama = args.model.startswith("llama")
    if llama:
        cfg = {"llama1b": LlamaConfig.llama_1b,
               "llama8b": LlamaConfig.llama3_8b}[args.model]()
    else:
        cfg = GPT2Config.tiny() if args.model == "tiny" else GPT2Config.small()
    if args.seq:
        cfg.block_size = min(cfg.block_size, args.seq) if args.model == "tiny" else args.seq
    fa = os.environ.get("SHTENS_FA", "")
    if fa and device.type == "cuda":
        try:
            torch.backends.cuda.preferred_rocm_fa_library(fa)
            log(f"rocm flash-attention backend: {fa}")
        except Exception as e:
            log(f"fa backend {fa} unavailable: {e}")
    blas = os.environ.get("SHTENS_BLAS", "")
    if blas and device.type == "cuda":
        try:
            torch.backends.cuda.preferred_blas_library(blas)
            log(f"blas library: {blas}")
        except Exception as e:
            log(f"blas {blas} unavailable: {e}")
    sdpa = os.environ.get("SHTENS_SDPA", "")
    if sdpa and device.type == "cuda":
        import sharedtensor_amd.models.gpt2 as _g
        from torch.nn.attention import SDPBackend, sdpa_kernel
        backend = {"flash": SDPBackend.FLASH_ATTENTION,
                   "mem_efficient": SDPBackend.EFFICIENT_ATTENTION,
                   "math": SDPBackend.MATH}[sdpa]
        _orig = _g.F.scaled_dot_product_attention

        def _sdpa(*a, **k):
            with sdpa_kernel(backend):
                return _orig(*a, **k)
        _g.F.scaled_dot_product_attention = _sdpa
        log(f"sdpa backend: {sdpa}")
    torch.manual_seed(1234)  # same random init on every rank
    model = (Llama(cfg) if llama else GPT2(cfg)).to(device)
    log(f"model {args.model}: {model.num_params()/1e6:.1f}M params, device {device}")

    use_bf16_params = device.type == "cuda" and not args.fp32_params
    trainer = AsyncDPTrainer(
        model, host="127.0.0.1",
        port_base=int(os.environ.get("SHTENS_PORT_BASE", 21000)),
        rank=rank, world=world, lr=args.lr, momentum=0.9,
        optimizer=args.opt, weight_decay=0.01 if args.opt == "adamw" else 0.0,
        amp_dtype=torch.bfloat16 if device.type == "cuda" else None,
        param_dtype=torch.bfloat16 if use_bf16_params else torch.float32,
        codec=args.codec, use_rccl=not args.no_rccl,
        lagged_scale=(not args.exact_scale) and device.type == "cuda",
        use_graphs=args.graphs,
        delta_dtype=torch.bfloat16 if args.bf16_deltas else torch.float32,
        sync_interval_s=args.sync_interval,
        snapshot_join=True)

    B, T = args.batch, cfg.block_size
    gen = torch.Generator(device="cpu").manual_seed(42 + rank)
    batches = [torch.randint(0, cfg.vocab_size, (B, T + 1), generator=gen).to(device)
               for _ in range(4)]

    dist = dist_setup(world)

    def one_step(i):
        b = batches[i % len(batches)]
        loss = trainer.step(b[:, :-1], b[:, 1:])
        return loss

    if device.type == "cuda":
        # clock warm: fresh leases start in a low-power state and the DVFS
        # ramp takes ~1-2 s — longer than a few warmup steps; burn a bounded
        # 2 s of GEMM before the (untimed) warmup so the timed region runs
        # at steady clocks
        wa = torch.randn(4096, 4096, device=device, dtype=torch.bfloat16)
        t_warm = time.perf_counter()
        while time.perf_counter() - t_warm < 2.0:
            wa = wa @ wa
            wa = wa / wa.norm().clamp_min(1e-6)
        torch.cuda.synchronize()
    log(f"warmup {args.warmup} steps")
    for i in range(args.warmup):
        one_step(i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    barrier(dist)
    if device.type == "cuda":
        torch.cuda.synchronize()
    s0 = trainer.stats()
    t0 = time.perf_counter()
    last_loss = None
    for i in range(args.steps):
        last_loss = one_step(args.warmup + i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    barrier(dist)
    if device.type == "cuda":
        torch.cuda.synchronize()
    s1 = trainer.stats()
    dt = max_over_ranks(dist, t1 - t0)

    wire_bytes = sum_over_ranks(
        dist, float((s1["bytes_sent"] - s0["bytes_sent"]) +
                    (s1["bytes_recv"] - s0["bytes_recv"])))
    rounds = sum_over_ranks(
        dist, float((s1["rounds_sent"] - s0["rounds_sent"])))
    n_params = trainer.shared.n
    tokens = float(B * T * args.steps * world)
    value = tokens / dt
    result = {
        "metric": ("async-DP tokens/sec GPT-2-small" if args.model in ("small", "tiny")
                   else f"async-DP tokens/sec {args.model}"),
        "value": round(value, 1),
        "unit": "tokens/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # reference publishes no numbers
        "dtype": "bf16" if device.type == "cuda" else "fp32",
        "data": "synthetic",
        "config": {
            "model": "gpt2-small" if args.model == "small" else args.model,
            "global_batch": B * world,
            "seq_len": T,
            "parallelism": f"async-dp{world}",
            "codec": args.codec,
            "n_params": n_params,
            "loss": round(float(last_loss), 4) if last_loss is not None else None,
            "paramsync_wire_gbps": round(wire_bytes / dt / 1e9, 3),
            "paramsync_logical_gbps": round(rounds * n_params * 4 / dt / 1e9, 3),
            "sync_rounds_per_s": round(rounds / dt, 1),
            "staleness_p50": s1["staleness_p50"],
            "staleness_p90": s1["staleness_p90"],
        },
    }
    if world == 1:
        # self-describing: a 1-rank tree has no links, so the param-sync half
        # of the metric does not exist rather than measuring zero
        result["config"]["paramsync"] = "n/a at n_gpus=1 (no links)"
        for k in ("paramsync_wire_gbps", "paramsync_logical_gbps",
                  "sync_rounds_per_s"):
            result["config"][k] = None
    else:
        # rank 0's per-link view (root of the tree): rounds/s per link and
        # whether the data plane upgraded to RCCL/xGMI
        result["config"]["links_rank0"] = [
            {"peer": l1["peer"], "rccl": l1["rccl"],
             "rounds_sent_per_s": round((l1["rounds_sent"] - l0["rounds_sent"]) / dt, 1),
             "rounds_recv_per_s": round((l1["rounds_recv"] - l0["rounds_recv"]) / dt, 1)}
            for l0, l1 in zip(s0["links"], s1["links"]) if l1["active"]]
    trainer.close()
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()
    if rank == 0:
        emit(result)


def run_table(args, rank, world, device):
    """Table-of-tensors sync on a Llama model's parameters (BASELINE
    config 4: per-tensor scales)."""
    from sharedtensor_amd.engine import SharedTable
    from sharedtensor_amd.models.llama import Llama, LlamaConfig
    from sharedtensor_amd.parallel.async_dp import tree_children, tree_parent

    lcfg = {"tiny": LlamaConfig.tiny, "llama1b": LlamaConfig.llama_1b,
            "llama8b": LlamaConfig.llama3_8b}.get(args.model, LlamaConfig.llama_1b)()
    torch.manual_seed(7)
    model = Llama(lcfg).to(device)
    log(f"llama table: {model.num_params()/1e6:.0f}M params, "
        f"{len(list(model.parameters()))} tensors")
    port_base = int(os.environ.get("SHTENS_PORT_BASE", 21000))
    nchild = len(tree_children(rank, world))
    tj0 = time.perf_counter()
    sh = SharedTable(
        "127.0.0.1", port_base, {n: p.data for n, p in model.named_parameters()},
        codec=args.codec,
        lagged_scale=(not args.exact_scale) and device.type == "cuda",
        use_graphs=args.graphs,
        use_rccl=not args.no_rccl, expected_children=nchild if world > 1 else 0,
        provision_up=rank > 0,
        explicit_parent=f"127.0.0.1:{port_base + tree_parent(rank)}" if rank else "",
        listen_port=port_base + rank if world > 1 else 0)
    join_s = time.perf_counter() - tj0  # ctor includes join + snapshot
    # the table holds its own replica; free the model to fit 8B-scale runs
    # (2 x ~100 GB working set on one 288 GB device)
    del model
    if device.type == "cuda":
        torch.cuda.empty_cache()
    dist = dist_setup(world)
    n = sh.n
    delta = torch.randn(n, dtype=torch.float32, device=device) * 0.001

    def one_step():
        sh.add_from_tensors(delta)
        if device.type == "cuda":
            torch.cuda.synchronize()
        time.sleep(args.interval)

    for _ in range(args.warmup):
        one_step()
    barrier(dist)
    s0 = sh.stats()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    time.sleep(1.0)
    t1 = time.perf_counter()
    barrier(dist)
    s1 = sh.stats()
    dt = max_over_ranks(dist, t1 - t0)
    wire = sum_over_ranks(dist, float(s1["bytes_sent"] - s0["bytes_sent"] +
                                      s1["bytes_recv"] - s0["bytes_recv"]))
    rounds = sum_over_ranks(dist, float(s1["rounds_sent"] - s0["rounds_sent"]))
    result = {
        "metric": "table-of-tensors param-sync GB/s (llama)",
        "value": round(rounds * n * 4 / dt / 1e9, 3),
        "unit": "GB/s logical",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {
            "model": args.model, "numel": n,
            "n_tensors": len(sh.names), "codec": args.codec,
            "wire_gbps": round(wire / dt / 1e9, 3),
            "staleness_p50": s1["staleness_p50"],
            # worst rank's table create incl. join + snapshot (master: alloc)
            "join_s": round(max_over_ranks(dist, join_s if rank else 0.0), 2),
        },
    }
    sh.close()
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()
    if rank == 0:
        emit(result)


def run_paramsync(args, rank, world, device):
    """Secondary mode: raw param-sync bandwidth on a flat tensor
    (BASELINE configs 2 and 5)."""
    from sharedtensor_amd.engine import SharedFlat
    from sharedtensor_amd.parallel.async_dp import tree_children, tree_parent

    n = args.numel
    port_base = int(os.environ.get("SHTENS_PORT_BASE", 21000))
    nchild = len(tree_children(rank, world))
    sh = SharedFlat(
        "127.0.0.1", port_base, [n], device=device, codec=args.codec,
        lagged_scale=(not args.exact_scale) and device.type == "cuda",
        use_graphs=args.graphs,
        delta_dtype=torch.bfloat16 if args.bf16_deltas else torch.float32,
        use_rccl=not args.no_rccl, expected_children=nchild if world > 1 else 0,
        provision_up=rank > 0,
        explicit_parent=f"127.0.0.1:{port_base + tree_parent(rank)}" if rank else "",
        listen_port=port_base + rank if world > 1 else 0)
    tj0 = time.perf_counter()
    sh._start()  # non-masters: join walk + full snapshot state transfer
    join_s = time.perf_counter() - tj0
    dist = dist_setup(world)
    delta = torch.randn(n, dtype=torch.float32, device=device) * 0.01

    def one_step():
        sh._add_flat(delta)
        if device.type == "cuda":
            torch.cuda.synchronize()
        time.sleep(args.interval)

    for _ in range(args.warmup):
        one_step()
    barrier(dist)
    s0 = sh.stats()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    # let the engine drain for a fixed window so we measure sync, not add
    time.sleep(1.0)
    t1 = time.perf_counter()
    barrier(dist)
    s1 = sh.stats()
    dt = max_over_ranks(dist, t1 - t0)
    wire = sum_over_ranks(dist, float(s1["bytes_sent"] - s0["bytes_sent"] +
                                      s1["bytes_recv"] - s0["bytes_recv"]))
    rounds = sum_over_ranks(dist, float(s1["rounds_sent"] - s0["rounds_sent"]))
    result = {
        "metric": "param-sync GB/s (flat tensor)",
        "value": round(rounds * n * 4 / dt / 1e9, 3),
        "unit": "GB/s logical",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {
            "numel": n, "codec": args.codec,
            "wire_gbps": round(wire / dt / 1e9, 3),
            "staleness_p50": s1["staleness_p50"],
            # worst rank's join incl. snapshot state transfer (master ~0)
            "join_s": round(max_over_ranks(dist, join_s if rank else 0.0), 2),
        },
    }
    sh.close()
    if dist is not None:
        dist.barrier()
        dist.destroy_process_group()
    if rank == 0:
        emit(result)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--lr", type=float, default=0.01)
    ap.add_argument("--codec", choices=["1bit", "fp8", "int4"], default="1bit")
    ap.add_argument("--opt", choices=["sgd", "adamw"], default="sgd",
                    help="fused optimizer feeding the shared tensor")
    ap.add_argument("--model", choices=["small", "tiny", "llama1b", "llama8b"], default="small")
    ap.add_argument("--mode", choices=["train", "paramsync", "table"], default="train")
    ap.add_argument("--numel", type=int, default=268_435_456)  # 1 GB fp32
    ap.add_argument("--interval", type=float, default=0.01)
    ap.add_argument("--sync-interval", type=float, default=0.05,
                    help="min seconds between a link's sync rounds (0 = "
                         "free-run like the reference; pacing keeps the codec "
                         "kernels from starving training compute of HBM)")
    ap.add_argument("--no-rccl", action="store_true")
    ap.add_argument("--fp32-params", action="store_true",
                    help="compute on fp32 replica views (default: bf16 shadow)")
    ap.add_argument("--bf16-deltas", action="store_true",
                    help="store link residual deltas in bf16 (halves their "
                         "HBM footprint; for 100GB-scale tensors)")
    ap.add_argument("--graphs", action="store_true",
                    help="capture sync rounds into hipGraphs (measured "
                         "slower on MI355X; see profiles/README.md)")
    ap.add_argument("--exact-scale", action="store_true",
                    help="per-round exact scale reduce (default: lagged, fused into quantize)")
    ap.add_argument("--device", default="auto")
    args = ap.parse_args()

    # A torchrun-style context is recognized by RANK (always set per rank);
    # a stray WORLD_SIZE in the parent environment without RANK must NOT
    # suppress self-launch (that is the round-1 hang, inverted)
    if "RANK" not in os.environ and args.gpus > 1:
        sys.exit(self_launch(args))

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus)) \
        if "RANK" in os.environ else args.gpus
    local = int(os.environ.get("LOCAL_RANK", rank))
    if args.device == "auto":
        # modulo device count so an N-rank rehearsal also runs on fewer GPUs
        device = torch.device(f"cuda:{local % torch.cuda.device_count()}") \
            if torch.cuda.is_available() else torch.device("cpu")
    else:
        device = torch.device(args.device)
    if device.type == "cuda":
        torch.cuda.set_device(device)
    if device.type == "cpu" and args.model == "small" and args.mode == "train":
        log("no GPU: downshifting to tiny model for a functional run")
        args.model = "tiny"
        args.batch = min(args.batch, 2)

    if args.mode == "train":
        run_train(args, rank, world, device)
    elif args.mode == "table":
        run_table(args, rank, world, device)
    else:
        run_paramsync(args, rank, world, device)


if __name__ == "__main__":
    main()
