#!/usr/bin/env python3
"""Codec kernel microbenchmark (GPU): reduce+quantize / apply throughput on a
flat fp32 tensor.  These are the engine's hot kernels; they should run at
HBM-bandwidth (MI355X: ~6.3 TB/s achievable of 8 TB/s peak).

Roofline per element (fp32 in):
  reduce:   4 B read                              (sampled: 4/stride)
  quantize: 4 B read + 4 B write + payload write  (1bit: +1/8 B)
  apply:    payload read + ndst * 8 B (atomic read-modify-write)
"""
import argparse
import json
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from sharedtensor_amd import _core  # noqa: E402


def time_kernel(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--numel", type=int, default=268_435_456)  # 1 GB fp32
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--stride", type=int, default=1)
    args = ap.parse_args()
    n = args.numel
    torch.cuda.set_device(0)
    s = torch.cuda.current_stream().cuda_stream
    results = {}
    for codec, name in [(0, "1bit"), (1, "fp8"), (2, "int4")]:
        d = torch.randn(n, device="cuda") * 0.01
        dc = _core.DevCodec(codec, [n], 0)
        scales = torch.zeros(1, dtype=torch.float32, device="cuda")
        pb = _core.payload_bytes(codec, n)
        payload = torch.zeros(pb, dtype=torch.uint8, device="cuda")
        v0 = torch.zeros(n, device="cuda")
        v1 = torch.zeros(n, device="cuda")
        v2 = torch.zeros(n, device="cuda")

        t_red = time_kernel(lambda: dc.reduce_scales(d.data_ptr(), scales.data_ptr(), args.stride, s), args.iters)
        # scale is fixed after the reduce; repeated quantize does identical
        # per-iteration work (residual oscillates +-scale)
        t_q = time_kernel(lambda: dc.quantize(d.data_ptr(), scales.data_ptr(), payload.data_ptr(), s), args.iters)
        t_ap1 = time_kernel(lambda: dc.apply(payload.data_ptr(), scales.data_ptr(), [v0.data_ptr()], s), args.iters)
        t_ap3 = time_kernel(lambda: dc.apply(payload.data_ptr(), scales.data_ptr(), [v0.data_ptr(), v1.data_ptr(), v2.data_ptr()], s), args.iters)
        results[name] = {
            "reduce_ms": round(t_red * 1e3, 3),
            "reduce_gbps": round(n * 4 / args.stride / t_red / 1e9, 1),
            "quantize_ms": round(t_q * 1e3, 3),
            "quantize_gbps": round((n * 8 + pb) / t_q / 1e9, 1),
            "apply1_ms": round(t_ap1 * 1e3, 3),
            "apply1_gbps_payload": round(pb / t_ap1 / 1e9, 1),
            "apply1_gbps_dst": round(n * 8 / t_ap1 / 1e9, 1),
            "apply3_ms": round(t_ap3 * 1e3, 3),
            "apply3_gbps_dst": round(n * 24 / t_ap3 / 1e9, 1),
            "payload_bytes": pb,
        }
        del d, payload, v0, v1, v2, dc
        torch.cuda.empty_cache()

    # fused sgd
    mom = torch.zeros(n, device="cuda")
    grad = torch.randn(n, device="cuda")
    vals = torch.zeros(n, device="cuda")
    d1 = torch.zeros(n, device="cuda")
    t_sgd = time_kernel(lambda: _core.gpu_fused_sgd(mom.data_ptr(), grad.data_ptr(), 0.1, 0.9, n,
                                                    [vals.data_ptr(), d1.data_ptr()], s), args.iters)
    results["fused_sgd_2dst"] = {"ms": round(t_sgd * 1e3, 3),
                                 "gbps": round(n * (8 + 4 + 16) / t_sgd / 1e9, 1)}
    print(json.dumps({"numel": n, "results": results}, indent=1))


if __name__ == "__main__":
    main()
