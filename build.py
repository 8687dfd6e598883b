#!/usr/bin/env python3
"""Build driver for the sharedtensor_amd native extension.

Compiles the C++ engine + CDNA4 HIP kernels with hipcc for gfx950 and links
them into an in-tree pybind11 module `sharedtensor_amd/_core.so`.  No CUDA
compat, no hipify, no torch headers — plain HIP + pybind11, so the module
loads on GPU-less build hosts (hipcc cross-compiles device code) and on
MI355X boxes alike.

Usage: python build.py [--force] [--debug]
"""
import argparse
import hashlib
import os
import subprocess
import sysconfig

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "sharedtensor_amd", "csrc")
OUT = os.path.join(ROOT, "sharedtensor_amd", "_core.so")
BUILD = os.path.join(ROOT, "build")

SOURCES = [
    "codec_cpu.cpp",
    "engine.cpp",
    "rccl_transport.cpp",
    "hip_kernels.hip",
    "ln_kernels.hip",
    "ce_kernels.hip",
    "gelu_kernels.hip",
    "swiglu_kernels.hip",
    "bindings.cpp",
]

HEADERS = ["common.h", "codec_cpu.h", "engine.h", "hip_api.h", "rccl_transport.h"]


def pybind11_include():
    import pybind11
    return pybind11.get_include()


def build(force=False, debug=False, verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    py_inc = sysconfig.get_paths()["include"]
    rocm = os.environ.get("ROCM_PATH", "/opt/rocm")
    hipcc = os.path.join(rocm, "bin", "hipcc")
    cxxflags = [
        "--offload-arch=gfx950",
        "-std=c++20",
        "-O3" if not debug else "-O0",
        "-g" if debug else "",
        "-fPIC",
        "-DSHAMD_WITH_HIP",
        f"-I{CSRC}",
        f"-I{py_inc}",
        f"-I{pybind11_include()}",
        f"-I{rocm}/include",
        "-Wno-unused-result",
        "-fvisibility=hidden",
        "-DNDEBUG" if not debug else "",
        "-parallel-jobs=4",
    ]
    cxxflags = [f for f in cxxflags if f]

    # content hash over sources+headers+flags for rebuild detection
    h = hashlib.sha256()
    for f in SOURCES + HEADERS:
        with open(os.path.join(CSRC, f), "rb") as fh:
            h.update(fh.read())
    h.update(" ".join(cxxflags).encode())
    stamp = os.path.join(BUILD, "core.stamp")
    digest = h.hexdigest()
    if not force and os.path.exists(OUT) and os.path.exists(stamp):
        with open(stamp) as fh:
            if fh.read().strip() == digest:
                if verbose:
                    print(f"[build.py] {OUT} up to date")
                return OUT

    objs = []
    procs = []
    for src in SOURCES:
        obj = os.path.join(BUILD, src.replace("/", "_") + ".o")
        objs.append(obj)
        cmd = [hipcc, "-c", os.path.join(CSRC, src), "-o", obj] + cxxflags
        if verbose:
            print("[build.py]", " ".join(cmd))
        procs.append(subprocess.Popen(cmd))
    rc = 0
    for p in procs:
        rc |= p.wait()
    if rc:
        raise RuntimeError("hipcc compilation failed")

    link = [hipcc, "-shared", "-fPIC", "-o", OUT] + objs + [
        f"-L{rocm}/lib", "-lamdhip64", "-lrccl", "-pthread",
        f"-Wl,-rpath,{rocm}/lib",
    ]
    if verbose:
        print("[build.py]", " ".join(link))
    subprocess.check_call(link)
    # guard against silently-unresolved engine symbols (python extensions
    # link with undefined symbols allowed; a stale signature would only
    # crash at first GPU call otherwise)
    nm = subprocess.run(["nm", "-C", "-u", OUT], capture_output=True, text=True)
    bad = [l for l in nm.stdout.splitlines() if "shamd::" in l]
    if bad:
        raise RuntimeError("unresolved engine symbols:\n" + "\n".join(bad))
    with open(stamp, "w") as fh:
        fh.write(digest)
    if verbose:
        print(f"[build.py] built {OUT}")
    return OUT


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--force", action="store_true")
    ap.add_argument("--debug", action="store_true")
    args = ap.parse_args()
    build(force=args.force, debug=args.debug)
