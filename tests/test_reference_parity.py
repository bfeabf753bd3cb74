"""Reference-parity: run the ACTUAL reference library (mounted read-only
at /root/reference) and dampr_amd's host engine on identical pipelines
and compare outputs.  This is the executable proof of API/semantics
parity — the reference is imported and executed, never copied.

Skipped where the reference tree is not mounted.
"""
import os
import sys

import pytest

REF = "/root/reference"
pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "dampr")),
    reason="reference tree not mounted")


def _load_reference():
    sys.path.insert(0, REF)
    import importlib
    for m in [m for m in list(sys.modules)
              if m == "dampr" or m.startswith("dampr.")]:
        del sys.modules[m]
    ref = importlib.import_module("dampr")
    sys.path.pop(0)
    return ref


@pytest.fixture(scope="module")
def engines():
    ref = _load_reference()
    from dampr_amd import Dampr as Ours
    return ref.Dampr, Ours


ITEMS = [("apple", 3), ("pear", 1), ("apple", 2), ("fig", 9),
         ("pear", 4), ("apple", 1), ("kiwi", 7)]

NUMS = list(range(40)) * 3


PIPELINES = {
    "count": lambda D: D.memory(NUMS).count(lambda x: x % 7),
    "fold_by": lambda D: D.memory(ITEMS).fold_by(
        lambda kv: kv[0], lambda a, b: a + b, value=lambda kv: kv[1]),
    "group_reduce": lambda D: D.memory(ITEMS)
        .group_by(lambda kv: kv[0], lambda kv: kv[1])
        .reduce(lambda _k, vs: sum(vs)),
    "sort_by": lambda D: D.memory(NUMS).sort_by(lambda v: -v),
    "unique": lambda D: D.memory(ITEMS)
        .group_by(lambda kv: kv[0], lambda kv: kv[1]).unique(),
    "topk": lambda D: D.memory(NUMS).topk(5),
    "len": lambda D: D.memory(NUMS).len(),
    "mean": lambda D: D.memory(ITEMS).mean(lambda kv: kv[0],
                                           lambda kv: kv[1]),
    "filter_flat": lambda D: D.memory(["a b", "b c c"])
        .flat_map(lambda s: s.split())
        .filter(lambda w: w != "a").count(),
    "a_group_by_sum": lambda D: D.memory(ITEMS)
        .a_group_by(lambda kv: kv[0], lambda kv: kv[1]).sum(),
    "join": lambda D: (
        D.memory(ITEMS).group_by(lambda kv: kv[0], lambda kv: kv[1])
        .join(D.memory(ITEMS).group_by(lambda kv: kv[0],
                                       lambda kv: kv[1]))
        .reduce(lambda l, r: sum(l) * sum(r))),
    "cross_right": lambda D: D.memory([1, 2, 3, 4])
        .cross_right(D.memory([10]), lambda v, t: v * t),
    "map_values": lambda D: D.memory([("a", ("x", 1)), ("b", ("y", 2))])
        .map_values(lambda v: v * 10),
    "map_keys": lambda D: D.memory([("a", ("x", 1)), ("b", ("y", 2))])
        .map_keys(lambda k: k.upper()),
    "prefix": lambda D: D.memory([3, 1, 2]).prefix(lambda v: v % 2),
    "suffix": lambda D: D.memory([3, 1, 2]).suffix(lambda v: v + 1),
    "inspect": lambda D: D.memory([1, 2]).inspect("t").count(),
    "cross_left": lambda D: D.memory([1, 2])
        .cross_left(D.memory([5, 7]), lambda a, b: (a, b)),
    "cross_set": lambda D: D.memory([1, 2, 3])
        .cross_set(D.memory([2, 3]), lambda v, right: v in right,
                   agg=set),
    # recognized (cross, agg) pair — the device engine folds the
    # broadcast side to one scalar; the reference executes its Python
    # MapAllJoin.  Parity pins the (streams-other, aggregates-self)
    # direction quirk too (SURVEY §8 note).
    "cross_set_sum": lambda D: D.memory([1, 2, 3])
        .cross_set(D.memory([4, 5]), __import__("operator").add,
                   agg=sum),
    # each key's 3 values are identical, so "first value seen" is
    # deterministic (arrival order within a key is scheduling-dependent
    # in both engines)
    "first": lambda D: D.memory(NUMS).a_group_by(lambda v: v).first(),
    "left_reduce": lambda D: (
        D.memory(ITEMS).group_by(lambda kv: kv[0], lambda kv: kv[1])
        .join(D.memory([("apple", 1), ("mango", 2)])
              .group_by(lambda kv: kv[0], lambda kv: kv[1]))
        .left_reduce(lambda l, r: (sorted(l), sorted(r)))),
    "checkpoint": lambda D: D.memory(NUMS)
        .map(lambda v: v + 1).checkpoint().count(lambda v: v % 5),
    "sample_all": lambda D: D.memory(NUMS).sample(1.0).len(),
    # joining two a_group_by().sum() outputs: the join values are the
    # keyed-reduce (k, v) tuples in BOTH engines (reference behavior --
    # tuple concatenation, not scalar addition)
    "agg_join": lambda D: (
        D.memory(ITEMS).a_group_by(lambda kv: kv[0], lambda kv: kv[1])
        .sum()
        .join(D.memory([("apple", 10), ("fig", 5)])
              .a_group_by(lambda kv: kv[0], lambda kv: kv[1]).sum())
        .reduce(lambda l, r: [a + b for a in l for b in r])),
    # join aggregates receive ONE-PASS group iterators (reference
    # semantics): a bare nested comprehension exhausts the right side
    # after the first left value -- only the first pair survives.  Both
    # our engines reproduce this for opaque lambdas (funcs.pair_* are
    # defined as the full product instead; see test_engine.py).
    # Each key's left values are IDENTICAL: "which value arrives first"
    # is scheduling-dependent in both engines, so distinct values made
    # this comparison flaky (2-of-4 failures round 1) while equal values
    # still pin the only-first-pair-survives semantics.
    "groupby_join_lazy": lambda D: (
        D.memory([("apple", 2), ("pear", 1), ("apple", 2), ("fig", 9),
                  ("apple", 2), ("fig", 9)])
        .group_by(lambda kv: kv[0], lambda kv: kv[1])
        .join(D.memory([("apple", 10), ("fig", 5)])
              .group_by(lambda kv: kv[0], lambda kv: kv[1]))
        .reduce(lambda l, r: [a + b for a in l for b in r], many=True)),
}


def _normalize(name, rows):
    if name == "unique":
        # first-seen value order within a group depends on worker/chunk
        # interleaving in BOTH engines (merge ties broken by run order):
        # compare as sets per group
        rows = [(k, tuple(sorted(v))) if isinstance(v, list) else (k, v)
                for k, v in rows]
    return sorted(map(repr, rows))


@pytest.mark.parametrize("name", sorted(PIPELINES))
def test_reference_parity(name, engines):
    RefD, OursD = engines
    build = PIPELINES[name]
    want = _normalize(name, build(RefD).run().read())
    got = _normalize(name, build(OursD).run().read())
    assert want, "vacuous comparison"
    assert got == want, (name, got[:4], want[:4])


@pytest.mark.parametrize("name", sorted(PIPELINES))
def test_reference_parity_device_engine(name, engines):
    """Same pipelines through the device engine (TorchOps here; HipOps
    on GPU boxes) — fallback and columnar paths must both match the
    reference."""
    from dampr_amd.gpu.engine import GpuRunner
    RefD, OursD = engines
    build = PIPELINES[name]
    want = _normalize(name, build(RefD).run().read())
    got = _normalize(name,
                     build(OursD).run(runner=GpuRunner).read())
    assert want, "vacuous comparison"
    assert got == want, (name, got[:4], want[:4])


def test_parity_multi_output_shared_subgraph(engines, tmp_path):
    """Dampr.run with a shared checkpointed root (word-stats idiom)."""
    RefD, OursD = engines

    def build(D):
        words = D.memory(["a b", "b c b", "a"]) \
            .flat_map(lambda s: s.split())
        counts = words.count()
        total = counts.fold_by(lambda _wc: 1, lambda x, y: x + y,
                               value=lambda wc: wc[1])
        return D.run(counts, total)

    ref_counts, ref_total = build(RefD)
    our_counts, our_total = build(OursD)
    assert sorted(our_counts.read()) == sorted(ref_counts.read())
    assert sorted(our_total.read()) == sorted(ref_total.read())


def test_parity_text_input(engines, tmp_path):
    RefD, OursD = engines
    f = tmp_path / "corpus.txt"
    f.write_text("x y z\nz z y\n" * 50)
    want = sorted(RefD.text(str(f))
                  .flat_map(lambda l: l.split()).count().run().read())
    got = sorted(OursD.text(str(f))
                 .flat_map(lambda l: l.split()).count().run().read())
    assert got == want and want


def test_parity_json_input(engines, tmp_path):
    RefD, OursD = engines
    f = tmp_path / "data.json"
    f.write_text('{"a": 1}\n{"a": 2}\n{"a": 5}\n')
    want = sorted(RefD.json(str(f)).map(lambda d: d["a"] * 2)
                  .run().read())
    got = sorted(OursD.json(str(f)).map(lambda d: d["a"] * 2)
                 .run().read())
    assert got == want and want


def test_parity_sink(engines, tmp_path):
    RefD, OursD = engines
    rd = tmp_path / "ref"
    od = tmp_path / "ours"
    RefD.memory(list(range(20))).map(str).sink(str(rd)).run()
    OursD.memory(list(range(20))).map(str).sink(str(od)).run()

    def lines(d):
        out = []
        for fn in sorted(os.listdir(d)):
            with open(os.path.join(d, fn)) as fh:
                out.extend(ln.strip() for ln in fh if ln.strip())
        return sorted(out)

    assert lines(od) == lines(rd) and lines(od)


def test_parity_partition_map_reduce(engines):
    RefD, OursD = engines

    def build(D):
        def pm(values):
            m = 0
            for v in values:
                m = max(m, v)
            yield 1, m

        def pr(groups):
            best = 0
            for _k, vs in groups:
                for v in vs:
                    best = max(best, v)
            yield 1, best

        return D.memory(list(range(100))) \
            .partition_map(pm).partition_reduce(pr)

    want = sorted(map(repr, build(RefD).run().read()))
    got = sorted(map(repr, build(OursD).run().read()))
    assert got == want and want


def test_parity_cached(engines):
    RefD, OursD = engines

    def build(D):
        base = D.memory(list(range(30))).map(lambda v: v * 2).cached()
        return base.count(lambda v: v % 3)

    want = sorted(build(RefD).run().read())
    got = sorted(build(OursD).run().read())
    assert got == want and want


def test_parity_filter_by_count(engines):
    RefD, OursD = engines
    sys.path.insert(0, REF)
    from dampr.utils import filter_by_count as ref_fbc
    sys.path.pop(0)
    from dampr_amd.utils import filter_by_count as our_fbc
    data = ["a", "b", "a", "c", "a", "b"]
    want = sorted(ref_fbc(RefD.memory(data), lambda x: x,
                          lambda c: c >= 2).read())
    got = sorted(our_fbc(OursD.memory(data), lambda x: x,
                         lambda c: c >= 2).read())
    assert got == want and want


def test_parity_custom_mapper_reducer(engines):
    """custom_mapper/custom_reducer take each library's own Map/Reduce
    operator objects — built per-engine, compared on output."""
    RefD, OursD = engines
    sys.path.insert(0, REF)
    from dampr.base import Map as RefMap, Reduce as RefReduce
    sys.path.pop(0)
    from dampr_amd.base import Map as OurMap, Reduce as OurReduce

    def build(D, Map, Reduce):
        return D.memory(list(range(20))) \
            .custom_mapper(Map(lambda _k, v: [(v % 3, v)])) \
            .custom_reducer(Reduce(lambda k, vs: sum(vs)))

    want = sorted(map(repr, build(RefD, RefMap, RefReduce).run().read()))
    got = sorted(map(repr, build(OursD, OurMap, OurReduce).run().read()))
    assert got == want and want


def test_parity_sink_tsv_json(engines, tmp_path):
    RefD, OursD = engines
    rows = [("hank", 755, 2.5), ("babe", 714, 3.5)]
    objs = [{"name": "hank", "hr": 755}, {"name": "babe", "hr": 714}]

    def lines(d):
        out = []
        for fn in sorted(os.listdir(d)):
            with open(os.path.join(d, fn)) as fh:
                out.extend(ln.rstrip("\n") for ln in fh if ln.strip())
        return sorted(out)

    RefD.memory(rows).sink_tsv(str(tmp_path / "rt")).run()
    OursD.memory(rows).sink_tsv(str(tmp_path / "ot")).run()
    assert lines(tmp_path / "ot") == lines(tmp_path / "rt") \
        and lines(tmp_path / "ot")
    RefD.memory(objs).sink_json(str(tmp_path / "rj")).run()
    OursD.memory(objs).sink_json(str(tmp_path / "oj")).run()
    assert sorted(lines(tmp_path / "oj")) == sorted(lines(tmp_path / "rj")) \
        and lines(tmp_path / "oj")


def test_parity_indexer(engines, tmp_path):
    RefD, OursD = engines
    f1 = tmp_path / "r" / "data.txt"
    f1.parent.mkdir()
    f1.write_text("apple red\nbanana yellow\ncherry red\n")
    f2 = tmp_path / "o" / "data.txt"
    f2.parent.mkdir()
    f2.write_text("apple red\nbanana yellow\ncherry red\n")
    sys.path.insert(0, REF)
    from dampr.utils import Indexer as RefIndexer
    sys.path.pop(0)
    from dampr_amd.utils.indexer import Indexer as OurIndexer
    ri = RefIndexer(str(f1))
    ri.build(lambda line: line.split())
    oi = OurIndexer(str(f2))
    oi.build(lambda line: line.split())
    want = sorted(ri.union(["red"]).read())
    got = sorted(oi.union(["red"]).read())
    assert got == want and want
    wanti = sorted(ri.intersect(["red", "apple"]).read())
    goti = sorted(oi.intersect(["red", "apple"]).read())
    assert goti == wanti and wanti
