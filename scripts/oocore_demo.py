"""Out-of-core demonstration (BASELINE config-5 direction): a group-by
over a synthetic (key, value) column FAR larger than the HBM pool,
forced through all three tiers (HBM -> pinned host -> NVMe), with exact
verification and per-tier byte counters.

The input is generated on device in chunks; lazy ingest admits batched
runs to the bounded pool, which evicts down the tiers as pressure
builds; the reduce pages partitions back.  Each record: key = row % K,
val = key, so group sums are exactly key * (rows / K) — verified.

Run on a GPU box:
  python scripts/oocore_demo.py --gb 100 --pool-gb 24 --host-gb 16
Env: DAMPR_SYNC_SPILL=1 disables the async D2H spill stream (A/B).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch

from dampr_amd import Dampr, funcs
from dampr_amd.gpu.engine import ColumnSource, GpuRunner


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gb", type=float, default=100.0)
    ap.add_argument("--pool-gb", type=float, default=24.0)
    ap.add_argument("--host-gb", type=float, default=16.0)
    ap.add_argument("--groups", type=int, default=1_000_000)
    ap.add_argument("--spill-dir", default="/tmp/dampr_oocore")
    args = ap.parse_args()
    dev = torch.device("cuda:0")

    rows = int(args.gb * (1 << 30)) // 16       # 8B key + 8B val
    K = args.groups
    rows -= rows % K                            # exact divisibility
    per_key = rows // K

    # device-side generation (chunked arange % K)
    t0 = time.perf_counter()
    keys = torch.empty(rows, dtype=torch.int64, device=dev)
    step = 1 << 27
    for lo in range(0, rows, step):
        hi = min(lo + step, rows)
        keys[lo:hi] = torch.arange(lo, hi, dtype=torch.int64,
                                   device=dev) % K
    vals = keys                                  # val = key (no copy)
    gen_s = time.perf_counter() - t0

    src = ColumnSource(keys, vals)
    from dampr_amd.runner import Graph
    from dampr_amd.dampr import PMap, Dampr as D
    source, ng = Graph().add_input(src)
    pipe = PMap(source, D(ng)).fold_by(funcs.identity, funcs.add)

    torch.cuda.synchronize()
    t1 = time.perf_counter()
    runner = GpuRunner("oocore", pipe.pmer.graph, n_partitions=64,
                       hbm_bytes=int(args.pool_gb * (1 << 30)),
                       host_bytes=int(args.host_gb * (1 << 30)),
                       spill_dir=args.spill_dir)
    if os.environ.get("DAMPR_SYNC_SPILL"):
        runner.pool.spill_stream = None
    ds = runner.run([pipe.source])[0]
    uk, agg = ds.columns()
    torch.cuda.synchronize()
    wall = time.perf_counter() - t1

    # exact verification: K groups, each sum = key * per_key
    n_groups = uk.numel()
    ok = bool(n_groups == K)
    if ok:
        want = uk.to(torch.int64) * per_key
        ok = bool(torch.equal(agg, want))

    stats = runner.stats
    print(json.dumps({
        "metric": "oocore_groupby_rows_per_sec",
        "value": rows / wall,
        "rows": rows,
        "gb_input": rows * 16 / (1 << 30),
        "pool_gb": args.pool_gb,
        "host_gb": args.host_gb,
        "groups": n_groups,
        "wall_s": wall,
        "gen_s": gen_s,
        "verified_exact": ok,
        "sync_spill": bool(os.environ.get("DAMPR_SYNC_SPILL")),
        "tier_stats": stats,
    }))
    assert ok, "verification failed"


if __name__ == "__main__":
    main()
