"""The columnar device engine: DSL pipelines over typed columns that
execute as gfx950 kernels (radix sort, segmented reduce, hash join) with
HBM-resident data.  Runs on CPU too (pure-torch oracle backend) — same
code, same results.

Usage: python examples/columns_device.py
"""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import numpy as np

from dampr_amd import Dampr, funcs


def main():
    rng = np.random.default_rng(0)

    # group-by count of 10M values: device hash/sort/segmented-reduce path
    vals = rng.integers(0, 100_000, size=10_000_000)
    counts = Dampr.columns(vals).count().run("value-counts")
    print("distinct values:", len(counts.read()))

    # per-key sums with an explicit key column
    keys = rng.integers(0, 1000, size=1_000_000)
    amounts = rng.integers(1, 100, size=1_000_000)
    sums = Dampr.columns(amounts, keys=keys) \
        .fold_by(funcs.identity, funcs.add).run("sums")
    print("sum groups:", len(sums.read()))

    # device hash join: matched pairs emitted straight from the K8 kernel
    lk = rng.integers(0, 5000, size=500_000)
    lv = rng.integers(0, 10, size=500_000)
    rk = rng.integers(2500, 7500, size=5000)
    rv = rng.integers(0, 10, size=5000)
    joined = Dampr.columns(lv, keys=lk) \
        .join(Dampr.columns(rv, keys=rk)) \
        .reduce(funcs.pair_product, many=True).run("join")
    print("join output rows:", len(joined.read()))

    # top-k by natural order (device radix top-k)
    top = Dampr.columns(rng.standard_normal(1_000_000)).topk(5).run("topk")
    print("top-5:", top.read())

    # string keys dictionary-encode onto the device kernels
    words = ["w%d" % i for i in rng.integers(0, 500, size=200_000)]
    from dampr_amd.gpu.engine import GpuRunner
    wc = Dampr.memory(words).count().run("wc", runner=GpuRunner)
    print("distinct words:", len(wc.read()))

    # string KEY columns: dictionary-encoded at ingest, joined on device
    # across different vocabularies (union-dictionary remap)
    accounts = np.array(["acct%03d" % i
                         for i in rng.integers(0, 800, size=300_000)])
    balances = rng.integers(1, 1000, size=300_000)
    flagged = np.array(["acct%03d" % i for i in range(0, 1000, 7)])
    ones = np.ones(flagged.size, dtype=np.int64)
    hits = Dampr.columns(balances, keys=accounts) \
        .join(Dampr.columns(ones, keys=flagged)) \
        .reduce(funcs.pair_left, many=True).run("flagged")
    print("flagged-account records:", len(hits.read()))


if __name__ == "__main__":
    main()
