"""Property-based cross-engine equivalence: random columnar pipelines
must produce identical results on the host engine (multiprocess Python
operators) and the device engine (TorchOps here; HipOps on GPU boxes
through the same code path)."""
import numpy as np
import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings as hyp_settings, strategies as st

from dampr_amd import Dampr, funcs
from dampr_amd.runner import MTRunner


@hyp_settings(max_examples=25, deadline=None, derandomize=True)
@given(
    vals=st.lists(st.integers(min_value=-1000, max_value=1000),
                  min_size=1, max_size=300),
    op=st.sampled_from(["count", "sum", "min", "max", "first", "topk",
                        "sort"]),
)
def test_engines_agree(vals, op):
    arr = np.array(vals, dtype=np.int64)

    def pipeline(pm):
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
            return pm.topk(5)
        return pm.sort_by()

    dev = pipeline(Dampr.columns(arr)).run()
    # host engine over the same records ((i, v) like columns' row keys)
    host = pipeline(Dampr.memory(list(arr.tolist()))).run(
        runner=MTRunner, n_maps=2, n_reducers=2)
    got = sorted(dev.read())
    want = sorted(host.read())
    assert got == want, (op, got[:5], want[:5])


@hyp_settings(max_examples=15, deadline=None, derandomize=True)
@given(
    lk=st.lists(st.integers(min_value=0, max_value=20), min_size=1,
                max_size=60),
    rk=st.lists(st.integers(min_value=0, max_value=20), min_size=1,
                max_size=60),
)
def test_join_engines_agree(lk, rk):
    lka = np.array(lk, dtype=np.int64)
    rka = np.array(rk, dtype=np.int64)
    lva = np.arange(len(lk), dtype=np.int64)
    rva = np.arange(len(rk), dtype=np.int64) * 3

    dev = Dampr.columns(lva, keys=lka) \
        .join(Dampr.columns(rva, keys=rka)) \
        .reduce(funcs.pair_sum, many=True).run()
    got = sorted(dev.read())
    want = []
    for i, k in enumerate(lk):
        for j, k2 in enumerate(rk):
            if k == k2:
                want.append((k, int(lva[i] + rva[j])))
    assert got == sorted(want)


@hyp_settings(max_examples=20, deadline=None, derandomize=True)
@given(
    vals=st.lists(st.integers(min_value=-500, max_value=500),
                  min_size=1, max_size=200),
    mod=st.integers(min_value=1, max_value=9),
    thresh=st.integers(min_value=-500, max_value=500),
)
def test_composite_pipelines_agree(vals, mod, thresh):
    """filter -> map -> count composites: device engine (fallback + device
    stages) vs host engine."""
    arr = np.array(vals, dtype=np.int64)

    def build(pm):
        return pm.filter(lambda v: v > thresh) \
            .map(lambda v: v % mod).count()

    dev = sorted(build(Dampr.columns(arr)).run().read())
    host = sorted(build(Dampr.memory(arr.tolist()))
                  .run(runner=MTRunner, n_maps=2, n_reducers=2).read())
    assert dev == host


_WORDS = ["apple", "fig", "kiwi", "pear", "plum", "yam", "oat", "rye"]


@hyp_settings(max_examples=25, deadline=None, derandomize=True)
@given(
    recs=st.lists(
        st.tuples(st.sampled_from(_WORDS),
                  st.integers(min_value=-100, max_value=100)),
        min_size=1, max_size=120),
    op=st.sampled_from(["count", "sum", "min", "max"]),
)
def test_string_key_engines_agree(recs, op):
    """String-keyed group pipelines: device engine (dictionary encode)
    vs host engine."""
    def build(D):
        pm = D.memory(recs)
        if op == "count":
            return pm.count(lambda kv: kv[0])
        g = pm.a_group_by(lambda kv: kv[0], lambda kv: kv[1])
        if op == "sum":
            return g.sum()
        return g.reduce(min if op == "min" else max)

    from dampr_amd.gpu.engine import GpuRunner
    dev = sorted(map(repr, build(Dampr).run(runner=GpuRunner).read()))
    host = sorted(map(repr, build(Dampr).run(
        runner=MTRunner, n_maps=2, n_reducers=2).read()))
    assert dev == host


@hyp_settings(max_examples=20, deadline=None, derandomize=True)
@given(
    lk=st.lists(st.sampled_from(_WORDS), min_size=1, max_size=40),
    rk=st.lists(st.sampled_from(_WORDS), min_size=1, max_size=40),
)
def test_string_key_join_engines_agree(lk, rk):
    """String-keyed device joins (union-dictionary remap) vs the exact
    oracle."""
    lka = np.array(lk)
    rka = np.array(rk)
    lva = np.arange(len(lk), dtype=np.int64)
    rva = np.arange(len(rk), dtype=np.int64) * 3
    dev = Dampr.columns(lva, keys=lka) \
        .join(Dampr.columns(rva, keys=rka)) \
        .reduce(funcs.pair_sum, many=True).run()
    got = sorted(dev.read())
    want = []
    for i, k in enumerate(lk):
        for j, k2 in enumerate(rk):
            if k == k2:
                want.append((k, int(lva[i] + rva[j])))
    assert got == sorted(want)


@hyp_settings(max_examples=15, deadline=None, derandomize=True)
@given(
    lk=st.lists(st.integers(min_value=0, max_value=12), min_size=1,
                max_size=40),
    rk=st.lists(st.integers(min_value=0, max_value=12), min_size=1,
                max_size=40),
    side=st.sampled_from(["left", "right"]),
)
def test_string_value_join_engines_agree(lk, rk, side):
    """Var-len VALUE columns through the device join vs host records."""
    lka = np.array([str(k) for k in lk])
    rka = np.array([str(k) for k in rk])
    lva = np.array(["L{}_{}".format(k, i) for i, k in enumerate(lk)])
    rva = np.array(["R{}_{}".format(k, i) for i, k in enumerate(rk)])
    agg = funcs.pair_left if side == "left" else funcs.pair_right
    dev = sorted(
        Dampr.columns(lva, keys=lka)
        .join(Dampr.columns(rva, keys=rka))
        .reduce(agg, many=True).run().read())
    host = sorted(
        Dampr.memory(list(zip(lka.tolist(), lva.tolist())))
        .group_by(lambda kv: kv[0], lambda kv: kv[1])
        .join(Dampr.memory(list(zip(rka.tolist(), rva.tolist())))
              .group_by(lambda kv: kv[0], lambda kv: kv[1]))
        .reduce(agg, many=True)
        .run(runner=MTRunner, n_maps=2, n_reducers=2).read())
    assert dev == host, (dev[:4], host[:4])


@hyp_settings(max_examples=15, deadline=None, derandomize=True)
@given(
    a=st.lists(st.integers(min_value=-50, max_value=50), min_size=1,
               max_size=40),
    b=st.lists(st.integers(min_value=-50, max_value=50), min_size=1,
               max_size=6),
    opname=st.sampled_from(["add", "mul", "min", "max"]),
)
def test_cross_engines_agree(a, b, opname):
    """Device broadcast cross joins vs host MapCrossJoin, any op in
    funcs.CROSS_BINOPS."""
    import operator
    op = {"add": operator.add, "mul": operator.mul,
          "min": min, "max": max}[opname]
    dev = sorted(
        Dampr.columns(np.array(a, dtype=np.int64))
        .cross_left(Dampr.columns(np.array(b, dtype=np.int64)), op)
        .run().read())
    host = sorted(
        Dampr.memory(a).cross_left(Dampr.memory(b), op)
        .run(runner=MTRunner, n_maps=2, n_reducers=2).read())
    assert dev == host, (dev[:5], host[:5])
