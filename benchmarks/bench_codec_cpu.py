#!/usr/bin/env python3
"""CPU codec throughput: the reference processes each link with ONE thread
(sharedtensor.c:133-189); this engine's CPU codec is a chunked parallel-for
(codec_cpu.cpp cpu_pfor).  Measures quantize / apply GB/s per codec at 1
thread vs N threads on this machine — the CPU-replica peers of BASELINE
config 5 run exactly this path.

Usage: python benchmarks/bench_codec_cpu.py [--numel N] [--threads T]
"""
import argparse
import json
import os
import subprocess
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])


def measure(numel, iters):
    import torch
    from sharedtensor_amd import _core
    from sharedtensor_amd.ops.oracle import CODEC_NAMES

    out = {}
    for name, cid in CODEC_NAMES.items():
        delta = (torch.randn(numel) * 0.1).contiguous()
        payload = torch.zeros(_core.payload_bytes(cid, numel),
                              dtype=torch.uint8)
        dst = torch.zeros(numel)
        scale = _core.cpu_scale(cid, delta.data_ptr(), numel, 1)
        t0 = time.perf_counter()
        for _ in range(iters):
            _core.cpu_encode(cid, delta.data_ptr(), numel, scale,
                             payload.data_ptr())
        tq = (time.perf_counter() - t0) / iters
        t0 = time.perf_counter()
        for _ in range(iters):
            _core.cpu_apply(cid, payload.data_ptr(), numel, scale,
                            [dst.data_ptr()])
        ta = (time.perf_counter() - t0) / iters
        out[f"{name}_quantize_gbps"] = round(numel * 4 / tq / 1e9, 2)
        out[f"{name}_apply_gbps"] = round(numel * 4 / ta / 1e9, 2)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--numel", type=int, default=268_435_456)  # 1 GB fp32
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--threads", type=int, default=0,
                    help="0 = default cap (min(hw,16))")
    ap.add_argument("--_measure", action="store_true")
    args = ap.parse_args()
    if args._measure:
        print(json.dumps(measure(args.numel, args.iters)))
        return
    result = {"numel": args.numel, "logical_gb": round(args.numel * 4 / 1e9, 2)}
    for tag, threads in (("1_thread", 1), ("parallel", args.threads)):
        env = dict(os.environ)
        if threads:
            env["SHTENS_CPU_THREADS"] = str(threads)
        elif tag == "parallel":
            env.pop("SHTENS_CPU_THREADS", None)
        else:
            env["SHTENS_CPU_THREADS"] = "1"
        r = subprocess.run(
            [sys.executable, os.path.abspath(__file__), "--_measure",
             "--numel", str(args.numel), "--iters", str(args.iters)],
            capture_output=True, text=True, env=env, timeout=1200)
        if r.returncode != 0:
            print(r.stderr[-500:], file=sys.stderr)
            sys.exit(1)
        result[tag] = json.loads(r.stdout.strip().splitlines()[-1])
    print(json.dumps(result, indent=1))


if __name__ == "__main__":
    main()
