"""Radix-sort micro-benchmark (K2/K3): raw radix_sort_pairs throughput
plus effective bytes/s, for the rs_scatter LDS-binning ablation.

Run on a GPU box: python scripts/bench_sort.py [--rows 200000000]
Prints one JSON line per case.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import numpy as np
import torch


def bench_case(name, keys, iters=5, warmup=2):
    from dampr_amd.gpu.relational import radix_sort_pairs
    times = []
    for i in range(warmup + iters):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        sk, sp = radix_sort_pairs(keys)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if i >= warmup:
            times.append(dt)
        del sk, sp
    dt = min(times)
    n = keys.numel()
    # one pass moves key(8B) + payload(4B) in and out = 24 B; count the
    # active passes for the effective traffic figure
    print(json.dumps({"case": name, "rows": n, "ms": dt * 1e3,
                      "rows_per_s": n / dt,
                      "gb_per_s_per_pass_if_8": n * 24 / dt / 8 / 1e9}))


def bench_torch_sort(name, keys, iters=5, warmup=2):
    """Calibration baseline: torch.sort = rocPRIM device radix sort."""
    times = []
    for i in range(warmup + iters):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        sk, sp = torch.sort(keys, stable=True)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if i >= warmup:
            times.append(dt)
        del sk, sp
    dt = min(times)
    print(json.dumps({"case": name + "-torchsort", "rows": keys.numel(),
                      "ms": dt * 1e3, "rows_per_s": keys.numel() / dt}))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000_000)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    rng = np.random.default_rng(0)
    n = args.rows

    # uniform 64-bit (all 8 passes active)
    k64 = torch.from_numpy(
        rng.integers(0, 1 << 63, size=n, dtype=np.int64)).to(dev)
    bench_case("u64-uniform", k64)
    bench_torch_sort("u64-uniform", k64)
    del k64

    # 20-bit keys (dict ranks / group ids; 3 active passes)
    k20 = torch.from_numpy(
        rng.integers(0, 1 << 20, size=n, dtype=np.int64)).to(dev)
    bench_case("u20-groupids", k20)
    bench_torch_sort("u20-groupids", k20)
    del k20

    # Zipf-skewed small cardinality (1M groups)
    z = (rng.zipf(1.3, size=n) - 1) % 1_000_000
    kz = torch.from_numpy(z.astype(np.int64)).to(dev)
    bench_case("zipf-1M", kz)
    del kz


if __name__ == "__main__":
    main()
