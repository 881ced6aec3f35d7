#!/usr/bin/env python3
"""MFMA GEMM perf sweep on the attestation kernel (run on a GPU box).

Reports TF/s at several sizes using torch-allocated buffers, validates
against torch fp32 at the smallest size, and prints one JSON line.
"""

import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from k8s_cc_manager_amd.ops import attest


def bench_size(n: int, iters: int = 20) -> dict:
    a = torch.randn(n, n, device="cuda", dtype=torch.float32).bfloat16()
    bt = torch.randn(n, n, device="cuda", dtype=torch.float32).bfloat16()
    c = torch.empty(n, n, device="cuda", dtype=torch.float32)
    out = {"n": n}
    for which, name in ((0, "v128"), (1, "v256"), (2, "v256w"), (3, "fp8")):
        if which in (1, 2) and (n % 256 or n % 128):
            continue
        if which == 3:
            if n % 128:
                continue
            a8 = torch.randn(n, n, device="cuda").to(torch.float8_e4m3fn)
            bt8 = torch.randn(n, n, device="cuda").to(torch.float8_e4m3fn)
            run = lambda: attest.mfma_gemm_fp8(  # noqa: E731
                0, a8.data_ptr(), bt8.data_ptr(), c.data_ptr(), n, n, n
            )
            run()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                run()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / iters
            out["fp8_ms"] = round(dt * 1e3, 3)
            out["fp8_tflops"] = round(2.0 * n * n * n / dt / 1e12, 1)
            del a8, bt8
            continue
        run = lambda: attest.mfma_gemm_bf16_variant(  # noqa: E731
            0, a.data_ptr(), bt.data_ptr(), c.data_ptr(), n, n, n, which
        )
        run()  # warm + correctness
        torch.cuda.synchronize()
        if n <= 2048:
            ref = a.float() @ bt.float().t()
            err = (c - ref).abs().max().item()
            out[f"{name}_max_abs_err"] = round(err, 6)
        t0 = time.perf_counter()
        for _ in range(iters):
            run()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        out[f"{name}_ms"] = round(dt * 1e3, 3)
        out[f"{name}_tflops"] = round(2.0 * n * n * n / dt / 1e12, 1)
    del a, bt, c
    torch.cuda.empty_cache()
    return out


def main():
    sizes = [int(s) for s in sys.argv[1:]] or [1024, 2048, 4096, 8192]
    results = [bench_size(n) for n in sizes]
    print(json.dumps({"kernel": "mfma_gemm_bf16 A/B", "results": results}))


if __name__ == "__main__":
    main()
