#!/usr/bin/env python3
"""Interleaved A/B of separately-compiled kernel experiment builds.

Loads each variant .so via ctypes in ONE process, verifies each against
torch, then times rounds round-robin (within-probe interleave; builds
are separate so no co-compilation perturbation)."""

import ctypes
import json
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402


def bind(path):
    lib = ctypes.CDLL(path)
    fn = lib.cc_mfma_gemm_bf16_variant
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_int] + [ctypes.c_void_p] * 3 + [ctypes.c_int] * 4
    return fn


VARIANT_FLAGS = {
    "base": [],
    "pf": ["-DCC_EXP_PREFETCH_FIRST=1"],
    "lgkm": ["-DCC_EXP_LGKM=1"],
    "both": ["-DCC_EXP_PREFETCH_FIRST=1", "-DCC_EXP_LGKM=1"],
}


def build_variants():
    """Build the experiment .so set if missing (they are gitignored)."""
    import subprocess

    src = Path(__file__).resolve().parent.parent / "k8s_cc_manager_amd/ops/attest_kernels.hip"
    outdir = Path("expbuild")
    outdir.mkdir(exist_ok=True)
    for name, flags in VARIANT_FLAGS.items():
        out = outdir / f"libcc_{name}.so"
        if out.exists() and out.stat().st_mtime >= src.stat().st_mtime:
            continue
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-shared",
             "-fPIC", "-Wno-unused-value", *flags, str(src), "-o", str(out)],
            check=True,
        )


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 6
    iters = 10
    build_variants()
    variants = {}
    for so in sorted(Path("expbuild").glob("libcc_*.so")):
        variants[so.stem.replace("libcc_", "")] = bind(str(so))

    a = torch.randn(n, n, device="cuda").bfloat16()
    bt = torch.randn(n, n, device="cuda").bfloat16()
    c = torch.empty(n, n, device="cuda", dtype=torch.float32)

    # verify each variant at 1024 first
    va = a[:1024, :1024].contiguous()
    vb = bt[:1024, :1024].contiguous()
    vc = torch.empty(1024, 1024, device="cuda", dtype=torch.float32)
    ref = va.float() @ vb.float().t()
    for name, fn in variants.items():
        rc = fn(0, va.data_ptr(), vb.data_ptr(), vc.data_ptr(), 1024, 1024, 1024, 1)
        torch.cuda.synchronize()
        err = (vc - ref).abs().max().item()
        assert rc == 0 and err < 2e-1, f"{name}: rc={rc} err={err}"

    results = {name: [] for name in variants}
    for _ in range(rounds):
        for name, fn in variants.items():
            fn(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), n, n, n, 1)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                fn(0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), n, n, n, 1)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / iters
            results[name].append(2.0 * n**3 / dt / 1e12)
    out = {
        name: {"median_tf": round(statistics.median(v), 1), "max_tf": round(max(v), 1)}
        for name, v in results.items()
    }
    print(json.dumps({"n": n, "rounds": rounds, "results": out}))


if __name__ == "__main__":
    main()
