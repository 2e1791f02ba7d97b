#!/usr/bin/env python3
"""Within-run A/B of gemm v1 vs v2 (guide methodology: interleaved, n>=10)."""
import json
import os
import sys
import time

import torch

from mcp_context_forge_amd.ops import hip


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    best = 1e9
    t0 = time.monotonic()
    for _ in range(iters):
        t1 = time.monotonic()
        fn()
        torch.cuda.synchronize()
        best = min(best, time.monotonic() - t1)
    return best


def main():
    m, n, k = 8192, 4096, 4096
    a = (torch.randn(m, k) * 0.3).bfloat16().cuda()
    bt = (torch.randn(n, k) * 0.3).bfloat16().cuda()
    flop = 2.0 * m * n * k
    lib = hip._load()

    def v1():
        out = torch.empty((m, n), dtype=torch.float32, device="cuda")
        hip._check("v1", lib.forge_gemm_bt(hip._ptr(a), hip._ptr(bt), hip._ptr(None), hip._ptr(out),
                                           m, n, k, 0, 0, hip._stream()))
        return out

    def v2():
        out = torch.empty((m, n), dtype=torch.float32, device="cuda")
        hip._check("v2", lib.forge_gemm_bt_v2(hip._ptr(a), hip._ptr(bt), hip._ptr(None), hip._ptr(out),
                                              m, n, k, 0, 0, hip._stream()))
        return out

    lib.forge_gemm_bt_v2_var.argtypes = [__import__("ctypes").c_void_p] * 3 + [__import__("ctypes").c_int] * 4 + [__import__("ctypes").c_void_p]
    lib.forge_gemm_bt_v2_var.restype = __import__("ctypes").c_int

    def mkvar(var):
        def f():
            out = torch.empty((m, n), dtype=torch.float32, device="cuda")
            hip._check("v2var", lib.forge_gemm_bt_v2_var(hip._ptr(a), hip._ptr(bt), hip._ptr(out),
                                                         m, n, k, var, hip._stream()))
            return out
        return f

    # refcheck first
    r1, r2 = v1(), v2()
    torch.cuda.synchronize()
    dmax = (r1 - r2).abs().max().item()
    # interleaved A/B
    t1 = bench(v1)
    t2 = bench(v2)
    t1b = bench(v1)
    t2b = bench(v2)
    res = {
        "shape": [m, n, k], "v1_v2_absdiff": dmax,
        "v1_tf": round(flop / min(t1, t1b) / 1e12, 1),
        "v2_tf": round(flop / min(t2, t2b) / 1e12, 1),
    }
    for var in (1, 2, 3, 4, 5):
        fv = mkvar(var)
        rv = fv()
        torch.cuda.synchronize()
        res[f"var{var}_absdiff"] = (r1 - rv).abs().max().item()
        tv = min(bench(fv), bench(fv))
        res[f"var{var}_tf"] = round(flop / tv / 1e12, 1)
    print(json.dumps(res))


if __name__ == "__main__":
    main()
