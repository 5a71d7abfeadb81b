#!/usr/bin/env python3
"""A/B the hand-written 256²-tile 8-phase MFMA GEMM against the library
(hipBLASLt via torch.nn.functional.linear) on the engine's projection
shapes.  Run on the GPU box:

    python tools/gemm_bench.py [--iters 50] [--check]

Shapes are the Llama-3-8B prefill projections (K=4096/14336) at the
chunked-prefill M regime (budget 8192) plus decode-sized M for the
threshold choice, and the 70B TP=8 shards.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

SHAPES = [
    # (tag, M, N, K)
    ("8b qkv      ", 8192, 6144, 4096),
    ("8b o        ", 8192, 4096, 4096),
    ("8b gate_up  ", 8192, 28672, 4096),
    ("8b down     ", 8192, 4096, 14336),
    ("8b qkv  M2k ", 2048, 6144, 4096),
    ("8b gup  M2k ", 2048, 28672, 4096),
    ("8b down M2k ", 2048, 4096, 14336),
    ("8b gup  M1k ", 1024, 28672, 4096),
    ("8b gup  M512", 512, 28672, 4096),
    ("8b gup  M256", 256, 28672, 4096),
    ("lmhead M512 ", 512, 128256, 4096),
    ("70b qkv tp8 ", 8192, 1280, 8192),
    ("70b gup tp8 ", 8192, 7168, 8192),
    ("70b down tp8", 8192, 8192, 3584),
]


def timeit(fn, iters):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--only", default=None, help="substring filter on shape tag")
    args = ap.parse_args()

    from agentcontrolplane_amd.ops import hip

    torch.manual_seed(0)
    print(f"{'shape':14s} {'M':>5s} {'N':>6s} {'K':>6s}  {'lib µs':>8s} {'own µs':>8s}"
          f"  {'lib TF':>7s} {'own TF':>7s}  win")
    for tag, m, n, k in SHAPES:
        if args.only and args.only not in tag:
            continue
        x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        fl = 2.0 * m * n * k
        t_lib = timeit(lambda: torch.nn.functional.linear(x, w), args.iters)
        ours = n % 256 == 0 and k % 128 == 0
        if ours:
            t_own = timeit(lambda: hip.gemm_bf16(x, w), args.iters)
            if args.check:
                err = (hip.gemm_bf16(x, w).float() - x.float() @ w.float().t()).abs().max()
                assert err < 0.03 * k ** 0.5, (tag, float(err))
            win = "OWN" if t_own < t_lib else "lib"
            print(f"{tag:14s} {m:5d} {n:6d} {k:6d}  {t_lib*1e6:8.1f} {t_own*1e6:8.1f}"
                  f"  {fl/t_lib/1e12:7.1f} {fl/t_own/1e12:7.1f}  {win}")
        else:
            print(f"{tag:14s} {m:5d} {n:6d} {k:6d}  {t_lib*1e6:8.1f} {'—':>8s}"
                  f"  {fl/t_lib/1e12:7.1f} {'—':>7s}  lib (shape)")


if __name__ == "__main__":
    main()
