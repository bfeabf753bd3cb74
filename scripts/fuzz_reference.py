"""Reference-parity fuzz: random pipelines executed on the ACTUAL
reference library (read-only mount) and dampr_amd's host engine,
compared exactly.  Heavier than the committed parity suite; run ad hoc:

    python scripts/fuzz_reference.py [trials]
"""
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

REF = "/root/reference"


def load_reference():
    sys.path.insert(0, REF)
    import importlib
    for m in [m for m in list(sys.modules)
              if m == "dampr" or m.startswith("dampr.")]:
        del sys.modules[m]
    ref = importlib.import_module("dampr")
    sys.path.pop(0)
    return ref.Dampr


def main(trials=60, seed=99):
    RefD = load_reference()
    from dampr_amd import Dampr as OursD
    rng = random.Random(seed)
    words = ["ash", "oak", "elm", "fir", "yew", "box"]
    fails = 0
    for trial in range(trials):
        n = rng.randint(1, 60)
        kind = rng.choice(["ints", "pairs", "strs"])
        if kind == "ints":
            data = [rng.randint(-40, 40) for _ in range(n)]
        elif kind == "strs":
            data = [rng.choice(words) for _ in range(n)]
        else:
            data = [(rng.choice(words), rng.randint(-9, 9))
                    for _ in range(n)]
        mod = rng.randint(1, 6)
        k = rng.randint(1, 8)
        op = rng.choice(
            ["count", "sort", "topk", "len", "fold", "group_sum",
             "filter_count", "prefix_count", "chain",
             "join_sum", "left_join", "cross_right", "agg_join"])
        if op in ("join_sum", "left_join", "agg_join") \
                and kind == "ints":
            # int keys joined against string rdata keys would crash the
            # reference worker (its parent then deadlocks -- no fault
            # tolerance); keep join key types aligned
            op = "count"
        rdata = [(rng.choice(words), rng.randint(-9, 9))
                 for _ in range(rng.randint(1, 20))]

        def build(D, data=data, kind=kind, mod=mod, k=k, op=op,
                  rdata=rdata):
            pm = D.memory(data)
            keyf = ((lambda kv: kv[0]) if kind == "pairs"
                    else (lambda v: v))
            valf = ((lambda kv: kv[1]) if kind == "pairs"
                    else (lambda v: 1))
            if op == "count":
                return pm.count(keyf)
            if op == "sort":
                return pm.sort_by(lambda v: repr(v))
            if op == "topk":
                # natural tuple order of (key, count): deterministic
                return pm.count(keyf).topk(k)
            if op == "len":
                return pm.len()
            if op == "fold":
                return pm.fold_by(keyf, lambda a, b: a + b, value=valf)
            if op == "group_sum":
                return pm.group_by(keyf, valf) \
                    .reduce(lambda _k, vs: sum(vs))
            if op == "join_sum":
                kf = ((lambda kv: kv[0]) if kind == "pairs"
                      else (lambda v: v))
                vf = ((lambda kv: kv[1]) if kind == "pairs"
                      else (lambda v: 1))
                return pm.group_by(kf, vf).join(
                    D.memory(rdata).group_by(lambda kv: kv[0],
                                             lambda kv: kv[1])) \
                    .reduce(lambda l, r: (sorted(l), sorted(r)))
            if op == "left_join":
                kf = ((lambda kv: kv[0]) if kind == "pairs"
                      else (lambda v: v))
                return pm.group_by(kf).join(
                    D.memory(rdata).group_by(lambda kv: kv[0])) \
                    .left_reduce(lambda l, r: (sorted(map(repr, l)),
                                               sorted(map(repr, r))))
            if op == "cross_right":
                return pm.cross_right(
                    D.memory(rdata).fold_by(lambda _x: 1,
                                            lambda a, b: a + b,
                                            value=lambda kv: kv[1]),
                    lambda v, kv: (repr(v), kv[1]))
            if op == "agg_join":
                kf = ((lambda kv: kv[0]) if kind == "pairs"
                      else (lambda v: v))
                vf = ((lambda kv: kv[1]) if kind == "pairs"
                      else (lambda v: 1))
                return pm.a_group_by(kf, vf).sum().join(
                    D.memory(rdata)
                    .a_group_by(lambda kv: kv[0],
                                lambda kv: kv[1]).sum()) \
                    .reduce(lambda l, r: (sorted(map(repr, l)),
                                          sorted(map(repr, r))))
            if op == "filter_count":
                return pm.filter(lambda v: hash(repr(v)) % 2 == 0) \
                    .count(keyf)
            if op == "prefix_count":
                return pm.count(keyf).prefix(lambda kv: kv[1] % mod)
            return pm.map(lambda v: repr(v)).flat_map(list) \
                .filter(lambda c: c not in "(),' ").count()

        try:
            want = sorted(map(repr, build(RefD).run().read()))
        except Exception as e:          # noqa: BLE001
            print("REF CRASH", trial, op, kind, ":", e)
            continue
        got = sorted(map(repr, build(OursD).run().read()))
        if got != want:
            fails += 1
            print("PARITY MISMATCH", trial, op, kind, n)
            print("  ref :", want[:4])
            print("  ours:", got[:4])
            if fails > 5:
                break
        if trial and trial % 20 == 0:
            print("...", trial, "trials, fails:", fails)
    print("done:", trials, "trials, fails:", fails)
    return fails


if __name__ == "__main__":
    sys.exit(1 if main(int(sys.argv[1])
                       if len(sys.argv) > 1 else 60) else 0)
