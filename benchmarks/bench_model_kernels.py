#!/usr/bin/env python3
"""Microbenchmark the model-side fused kernels (LN, CE) vs torch equivalents
on the GPT-2-small B=64 shapes."""
import json
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from sharedtensor_amd import _core  # noqa: E402


def t(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    torch.cuda.set_device(0)
    s = torch.cuda.current_stream().cuda_stream
    out = {}

    # ---- LN (R=65536, C=768) ----
    R, C = 65536, 768
    x = torch.randn(R, C, device="cuda").to(torch.bfloat16)
    w = torch.randn(C, device="cuda").to(torch.bfloat16)
    b = torch.randn(C, device="cuda").to(torch.bfloat16)
    dy = torch.randn(R, C, device="cuda").to(torch.bfloat16)
    y = torch.empty_like(x)
    dx = torch.empty_like(x)
    mean = torch.empty(R, dtype=torch.float32, device="cuda")
    rstd = torch.empty(R, dtype=torch.float32, device="cuda")
    dwdb = torch.zeros(2 * C, dtype=torch.float32, device="cuda")
    out["ln_fwd_ms"] = round(t(lambda: _core.ln_fwd(
        x.data_ptr(), w.data_ptr(), b.data_ptr(), y.data_ptr(),
        mean.data_ptr(), rstd.data_ptr(), R, C, 1e-5, s)), 3)
    out["ln_bwd_ms"] = round(t(lambda: _core.ln_bwd(
        dy.data_ptr(), x.data_ptr(), w.data_ptr(), mean.data_ptr(),
        rstd.data_ptr(), dx.data_ptr(), dwdb.data_ptr(),
        dwdb[C:].data_ptr(), R, C, s)), 3)
    xt = x.clone().requires_grad_(True)
    wt = w.clone().requires_grad_(True)
    bt = b.clone().requires_grad_(True)
    out["torch_ln_fwd_ms"] = round(t(lambda: torch.nn.functional.layer_norm(
        xt, (C,), wt, bt, 1e-5)), 3)
    yt = torch.nn.functional.layer_norm(xt, (C,), wt, bt, 1e-5)
    out["torch_ln_bwd_ms"] = round(t(lambda: torch.autograd.grad(
        yt, (xt, wt, bt), dy, retain_graph=True)), 3)

    # ---- GELU (65536 x 3072) ----
    xg = torch.randn(65536, 3072, device="cuda").to(torch.bfloat16)
    yg = torch.empty_like(xg)
    dyg = torch.randn_like(xg)
    dxg = torch.empty_like(xg)
    ng = xg.numel()
    out["gelu_fwd_ms"] = round(t(lambda: _core.gelu_fwd(
        xg.data_ptr(), yg.data_ptr(), ng, s)), 3)
    out["gelu_bwd_ms"] = round(t(lambda: _core.gelu_bwd(
        dyg.data_ptr(), xg.data_ptr(), dxg.data_ptr(), ng, s)), 3)
    xt2 = xg.clone().requires_grad_(True)
    out["torch_gelu_fwd_ms"] = round(t(lambda: torch.nn.functional.gelu(
        xt2, approximate="tanh")), 3)
    yt2 = torch.nn.functional.gelu(xt2, approximate="tanh")
    out["torch_gelu_bwd_ms"] = round(t(lambda: torch.autograd.grad(
        yt2, xt2, dyg, retain_graph=True)), 3)

    # ---- RMSNorm (llama-1B shape R=B*T=16384, C=2048; 8B C=4096) ----
    for C2 in (2048, 4096):
        R2 = 16384
        xr = torch.randn(R2, C2, device="cuda").to(torch.bfloat16)
        wr = torch.randn(C2, device="cuda").to(torch.bfloat16)
        dyr = torch.randn(R2, C2, device="cuda").to(torch.bfloat16)
        yr = torch.empty_like(xr)
        dxr = torch.empty_like(xr)
        rstd2 = torch.empty(R2, dtype=torch.float32, device="cuda")
        dg = torch.zeros(C2, dtype=torch.float32, device="cuda")
        out[f"rms{C2}_fwd_ms"] = round(t(lambda: _core.rms_fwd(
            xr.data_ptr(), wr.data_ptr(), yr.data_ptr(), rstd2.data_ptr(),
            R2, C2, 1e-5, s)), 3)
        out[f"rms{C2}_bwd_ms"] = round(t(lambda: _core.rms_bwd(
            dyr.data_ptr(), xr.data_ptr(), wr.data_ptr(), rstd2.data_ptr(),
            dxr.data_ptr(), dg.data_ptr(), R2, C2, s)), 3)

        def torch_rms(xi, wi):  # the pre-fusion module path (fp32 upcast)
            xf = xi.float()
            h = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
            return (h * wi.float()).to(xi.dtype)
        xt3 = xr.clone().requires_grad_(True)
        wt3 = wr.clone().requires_grad_(True)
        out[f"torch_rms{C2}_fwd_ms"] = round(t(lambda: torch_rms(xt3, wt3)), 3)
        yt3 = torch_rms(xt3, wt3)
        out[f"torch_rms{C2}_bwd_ms"] = round(t(lambda: torch.autograd.grad(
            yt3, (xt3, wt3), dyr, retain_graph=True)), 3)

    # ---- bias-grad column sum (R=65536, C=768 and 3072) ----
    for Cc in (768, 3072):
        Rc = 65536
        xc = torch.randn(Rc, Cc, device="cuda").to(torch.bfloat16)
        oc = torch.zeros(Cc, dtype=torch.float32, device="cuda")
        out[f"colsum{Cc}_ms"] = round(t(lambda: _core.colsum_bf16(
            xc.data_ptr(), oc.data_ptr(), Rc, Cc, s)), 3)
        out[f"torch_colsum{Cc}_ms"] = round(t(lambda: xc.sum(0)), 3)

    # ---- SwiGLU (llama-1B MLP shape R=8192, F=8192) ----
    Rs, Fs = 8192, 8192
    sx1 = torch.randn(Rs, Fs, device="cuda").to(torch.bfloat16)
    sx3 = torch.randn(Rs, Fs, device="cuda").to(torch.bfloat16)
    sdy = torch.randn(Rs, Fs, device="cuda").to(torch.bfloat16)
    sy = torch.empty_like(sx1)
    sdx1 = torch.empty_like(sx1)
    sdx3 = torch.empty_like(sx3)
    ns = sx1.numel()
    out["swiglu_fwd_ms"] = round(t(lambda: _core.swiglu_fwd(
        sx1.data_ptr(), sx3.data_ptr(), sy.data_ptr(), ns, s)), 3)
    out["swiglu_bwd_ms"] = round(t(lambda: _core.swiglu_bwd(
        sdy.data_ptr(), sx1.data_ptr(), sx3.data_ptr(), sdx1.data_ptr(),
        sdx3.data_ptr(), ns, s)), 3)
    sa = sx1.clone().requires_grad_(True)
    sb = sx3.clone().requires_grad_(True)
    out["torch_swiglu_fwd_ms"] = round(t(
        lambda: torch.nn.functional.silu(sa) * sb), 3)
    syt = torch.nn.functional.silu(sa) * sb
    out["torch_swiglu_bwd_ms"] = round(t(lambda: torch.autograd.grad(
        syt, (sa, sb), sdy, retain_graph=True)), 3)

    # ---- CE (R=65536, V=50257) ----
    R, V = 65536, 50257
    logits = (torch.randn(R, V, device="cuda") * 2).to(torch.bfloat16)
    targets32 = torch.randint(0, V, (R,), device="cuda", dtype=torch.int32)
    targets = targets32.long()
    loss = torch.empty(R, dtype=torch.float32, device="cuda")
    row_m = torch.empty(R, dtype=torch.float32, device="cuda")
    row_lse = torch.empty(R, dtype=torch.float32, device="cuda")
    dlog = torch.empty_like(logits)
    g = torch.ones((), dtype=torch.float32, device="cuda")
    out["ce_fwd_ms"] = round(t(lambda: _core.ce_fwd(
        logits.data_ptr(), targets32.data_ptr(), loss.data_ptr(),
        row_m.data_ptr(), row_lse.data_ptr(), R, V, s), iters=5), 3)
    out["ce_bwd_ms"] = round(t(lambda: _core.ce_bwd(
        logits.data_ptr(), targets32.data_ptr(), row_lse.data_ptr(),
        dlog.data_ptr(), g.data_ptr(), 1.0 / R, R, V, s), iters=5), 3)
    lt = logits.clone().requires_grad_(True)
    out["torch_ce_fwd_ms"] = round(t(lambda: torch.nn.functional.cross_entropy(
        lt, targets), iters=5), 3)
    lt2 = logits.clone().requires_grad_(True)
    losst = torch.nn.functional.cross_entropy(lt2, targets)
    out["torch_ce_bwd_ms"] = round(t(lambda: torch.autograd.grad(
        losst, lt2, retain_graph=True), iters=5), 3)
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
