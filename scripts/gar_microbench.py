#!/usr/bin/env python3
"""Microbenchmark of the HIP GAR kernels on ResNet-50-sized gradients.

Times each GAR on an [n, d] fp32 matrix (default n=8, d=25.6M ~ ResNet-50)
resident in HBM, reports ms and achieved HBM bandwidth vs the algorithmic
minimum bytes, and compares against eager PyTorch (AGGREGATHOR_FORCE_EAGER
path) where an eager formulation exists.

Usage: python scripts/gar_microbench.py [--n 8] [--d 25600000] [--iters 50]
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def time_gpu(fn, iters, warmup=5):
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
    ap.add_argument("--n", type=int, default=8)
    ap.add_argument("--f", type=int, default=2)
    ap.add_argument("--d", type=int, default=25_600_000)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available()

    from aggregathor_amd import ops
    assert ops.hip_available()
    ext = ops._load_extension()

    n, d, f = args.n, args.d, args.f
    m = n - f - 2
    g = torch.randn((n, d), device="cuda", dtype=torch.float32)
    matrix_gb = n * d * 4 / 1e9

    results = {"n": n, "d": d, "f": f, "matrix_GB": round(matrix_gb, 3)}
    print(f"# GAR microbench: n={n} d={d} f={f} matrix={matrix_gb:.2f} GB")

    cases = [
        ("pairwise_sqdist", lambda: ext.pairwise_sqdist(g), n * d * 4),
        ("krum", lambda: ext.krum(g, f, m), (n + m + 0.25) * d * 4),
        ("median", lambda: ext.median(g), (n + 1) * d * 4),
        ("averaged_median", lambda: ext.averaged_median(g, n - f),
         (n + 1) * d * 4),
        ("average_nan", lambda: ext.average_nan(g), (n + 1) * d * 4),
    ]
    if n >= 4 * f + 3:
        cases.append(("bulyan", lambda: ext.bulyan(g, f, m),
                      (2 * n + 1) * d * 4))
    for name, fn, min_bytes in cases:
        dt = time_gpu(fn, args.iters)
        bw = min_bytes / dt / 1e12
        results[name] = {"ms": round(dt * 1e3, 3),
                         "algorithmic_TBps": round(bw, 2)}
        print(f"{name:20s} {dt * 1e3:8.3f} ms   {bw:6.2f} TB/s algorithmic")

    # Eager-PyTorch comparisons (same math, GPU eager ops).
    def eager_krum():
        dist = torch.cdist(g, g, p=2).pow_(2)
        dist.fill_diagonal_(float("inf"))
        scores = torch.sort(dist, dim=1).values[:, : n - f - 2].sum(dim=1)
        sel = torch.argsort(scores)[:m]
        return g[sel].mean(dim=0)

    def eager_median():
        return torch.nan_to_num(g, nan=float("inf")).median(dim=0).values

    for name, fn in [("eager_krum_cdist", eager_krum),
                     ("eager_median", eager_median)]:
        try:
            dt = time_gpu(fn, max(args.iters // 5, 3))
            results[name] = {"ms": round(dt * 1e3, 3)}
            print(f"{name:20s} {dt * 1e3:8.3f} ms   (eager PyTorch)")
        except RuntimeError as e:
            print(f"{name}: failed ({e})")

    print(json.dumps(results))


if __name__ == "__main__":
    main()
