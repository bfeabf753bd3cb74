"""Deep cross-engine fuzz: N random pipelines over random int columns,
device engine vs host engine, exact comparison.  Heavier than the
committed hypothesis suite (tests/test_property.py); run ad hoc:

    python scripts/fuzz_engines.py [trials]

Passed 500/500 at r1 close.
"""
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

from dampr_amd import Dampr, funcs
from dampr_amd.runner import MTRunner


def main(trials=500, seed=123):
    rng = random.Random(seed)
    fails = 0
    for trial in range(trials):
        n = rng.randint(1, 400)
        lo, hi = sorted(rng.sample(range(-2000, 2000), 2))
        vals = np.array([rng.randint(lo, hi) for _ in range(n)],
                        dtype=np.int64)
        op = rng.choice(["count", "sum", "min", "max", "first", "topk",
                         "sort", "len", "mean"])
        K = rng.randint(1, 12)

        def build(pm):
            if op == "count":
                return pm.count()
            if op == "sum":
                return pm.a_group_by().sum()
            if op == "min":
                return pm.a_group_by().reduce(min)
            if op == "max":
                return pm.a_group_by().reduce(max)
            if op == "first":
                return pm.a_group_by().first()
            if op == "topk":
                return pm.topk(K)
            if op == "len":
                return pm.len()
            if op == "mean":
                return pm.mean(funcs.identity)
            return pm.sort_by()

        dev = sorted(map(repr, build(Dampr.columns(vals)).run().read()))
        host = sorted(map(repr, build(Dampr.memory(vals.tolist())).run(
            runner=MTRunner, n_maps=2, n_reducers=2).read()))
        if dev != host:
            fails += 1
            print("MISMATCH", trial, op, n, K, dev[:3], host[:3])
            if fails > 5:
                break
        if trial and trial % 100 == 0:
            print("...", trial, "trials, fails:", fails)
    print("done:", trials, "trials, fails:", fails)
    return fails


if __name__ == "__main__":
    sys.exit(1 if main(int(sys.argv[1])
                       if len(sys.argv) > 1 else 500) else 0)
