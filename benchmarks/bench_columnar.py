"""Columnar-engine benchmarks for BASELINE configs 3 and 4:

* config 3 — a_group_by().sum() over (key, value) columns: device radix
  sort + segmented reduce (K2/K3/K5/K7) through the DSL.
* config 4 — reduce-side equi-join: device chained hash join (K8)
  through the DSL.

Run on a GPU box:  python benchmarks/bench_columnar.py [--rows 200000000]
Prints one JSON line per config (whole-GPU rows/s).
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

from dampr_amd import Dampr, funcs
from dampr_amd.gpu.engine import GpuRunner


def bench_group_sum_values(rows, card, device, iters, warmup):
    """Group-by-sum over a value column (fold_by(identity, add)):
    device path = partition + radix sort + segmented reduce."""
    rng = np.random.default_rng(0)
    vals = torch.from_numpy(rng.integers(0, card, size=rows))
    if device.startswith("cuda"):
        vals = vals.to(device)          # resident in HBM (like bench.py)
    times = []
    for i in range(warmup + iters):
        t0 = time.perf_counter()
        out = Dampr.columns(vals).fold_by(funcs.identity, funcs.add) \
            .run(device=device)
        k, v = out.dataset.columns()
        torch.cuda.synchronize() if device.startswith("cuda") else None
        dt = time.perf_counter() - t0
        if i >= warmup:
            times.append(dt)
        n_groups = k.numel()
    dt = min(times)
    return {"metric": "groupby_sum_rows_per_sec", "value": rows / dt,
            "unit": "rows/s", "rows": rows, "groups": int(n_groups),
            "ms": dt * 1000, "config": "baseline-3"}


def bench_join(rows_l, rows_r, card, device, iters, warmup):
    rng = np.random.default_rng(1)
    def col(a):
        t = torch.from_numpy(a)
        return t.to(device) if device.startswith("cuda") else t
    lk = col(rng.integers(0, card, size=rows_l))
    lv = col(rng.integers(0, 1000, size=rows_l))
    rk = col(rng.integers(0, card, size=rows_r))
    rv = col(rng.integers(0, 1000, size=rows_r))
    times = []
    for i in range(warmup + iters):
        t0 = time.perf_counter()
        out = Dampr.columns(lv, keys=lk).join(
            Dampr.columns(rv, keys=rk)) \
            .reduce(funcs.pair_sum, many=True).run(device=device)
        k, v = out.dataset.columns()
        torch.cuda.synchronize() if device.startswith("cuda") else None
        dt = time.perf_counter() - t0
        if i >= warmup:
            times.append(dt)
        n_out = k.numel()
    dt = min(times)
    return {"metric": "hash_join_rows_per_sec",
            "value": (rows_l + rows_r) / dt, "unit": "rows/s",
            "rows_l": rows_l, "rows_r": rows_r, "out_rows": int(n_out),
            "ms": dt * 1000, "config": "baseline-4"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000_000,
                    help="rows PER RANK (weak scaling at world > 1)")
    ap.add_argument("--card", type=int, default=1_000_000)
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--trace", action="store_true",
                    help="print per-stage timings of the last run")
    args = ap.parse_args()
    # multi-rank (BASELINE configs 3/4 name 8x MI355X): launch via
    # torch.distributed.run; rows are PER RANK and the engine exchanges
    # partitions over RCCL (one process per GPU)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import torch.distributed as dist
        lr = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(lr % max(torch.cuda.device_count(), 1))
        dist.init_process_group("nccl")
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    if world > 1:
        device = "cuda:{}".format(
            int(os.environ.get("LOCAL_RANK", "0"))
            % max(torch.cuda.device_count(), 1))
    if device == "cpu":
        args.rows = min(args.rows, 1_000_000)
    if world > 1:
        # engine contract: every rank passes the IDENTICAL logical
        # input and keeps its 1/world slice — scale the total so the
        # per-rank share stays at --rows (weak scaling)
        args.rows *= world

    from dampr_amd.utils.trace import get_trace
    r = bench_group_sum_values(args.rows, args.card, device, args.iters,
                               args.warmup)
    r["device"] = device
    r["world"] = world
    if world <= 1 or int(os.environ.get("RANK", "0")) == 0:
        print(json.dumps(r))
    if args.trace:
        print(get_trace().report())
    r = bench_join(args.rows, args.rows // 8, max(args.rows, 1), device,
                   args.iters, args.warmup)
    r["device"] = device
    r["world"] = world
    if world <= 1 or int(os.environ.get("RANK", "0")) == 0:
        print(json.dumps(r))
    if args.trace:
        print(get_trace().report())


if __name__ == "__main__":
    main()
