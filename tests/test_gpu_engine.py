"""GPU-side columnar-engine tests: the same DSL pipelines as
tests/test_engine.py executed on cuda:0 with the HipOps backend (gfx950
kernels), checked against exact Python oracles.  A HipOps/TorchOps
cross-check guards backend divergence."""
import collections
import os

import numpy as np
import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif("not __import__('torch').cuda.is_available()",
                       reason="needs an MI355X"),
]

from dampr_amd import Dampr, funcs  # noqa: E402


def _run_dev(pm, **kw):
    return pm.run(device="cuda:0", **kw)


def test_hip_backend_selected():
    from dampr_amd.gpu.backend import HipOps, ops_for
    assert isinstance(ops_for(torch.device("cuda:0")), HipOps)


def test_count_device():
    rng = np.random.default_rng(0)
    vals = rng.integers(0, 1000, size=1_000_000)
    got = dict(_run_dev(Dampr.columns(vals).count()).read())
    want = collections.Counter(int(v) for v in vals)
    assert got == dict(want)


def test_fold_by_minmax_device():
    rng = np.random.default_rng(1)
    vals = rng.integers(-(1 << 40), 1 << 40, size=200_000)
    keys = rng.integers(0, 5000, size=200_000)
    for binop, pyop in ((min, min), (max, max), (funcs.add, sum)):
        got = dict(_run_dev(
            Dampr.columns(vals, keys=keys)
            .a_group_by(funcs.identity).reduce(binop)).read())
        groups = collections.defaultdict(list)
        for v in vals:
            groups[int(v)].append(int(v))
        want = {k: pyop(vs) for k, vs in groups.items()}
        assert got == want


def test_first_device():
    vals = np.array([7, 7, 3, 3, 3, 11], dtype=np.int64)
    got = dict(_run_dev(Dampr.columns(vals).a_group_by().first()).read())
    assert got == {7: 7, 3: 3, 11: 11}


def test_float_sum_device():
    rng = np.random.default_rng(2)
    vals = np.round(rng.standard_normal(50_000), 3)
    keys = rng.integers(0, 100, size=50_000)
    # key by the row-key column is not expressible; group equal floats
    got = dict(_run_dev(
        Dampr.columns(vals, keys=keys)
        .fold_by(funcs.identity, funcs.add)).read())
    groups = collections.defaultdict(float)
    for v in vals:
        groups[float(v)] += float(v)
    assert len(got) == len(groups)
    for k, s in list(groups.items())[:500]:
        assert abs(got[k] - s) < 1e-9 * max(1.0, abs(s))


def test_join_device():
    rng = np.random.default_rng(3)
    lk = rng.integers(0, 2000, size=100_000)
    lv = rng.integers(0, 1000, size=100_000)
    rk = rng.integers(1000, 3000, size=5000)
    rv = rng.integers(0, 1000, size=5000)
    out = _run_dev(Dampr.columns(lv, keys=lk)
                   .join(Dampr.columns(rv, keys=rk))
                   .reduce(funcs.pair_sum, many=True))
    got = sorted(out.read())
    rmap = collections.defaultdict(list)
    for k, v in zip(rk, rv):
        rmap[int(k)].append(int(v))
    want = []
    for k, v in zip(lk, lv):
        for rvv in rmap.get(int(k), ()):
            want.append((int(k), int(v) + rvv))
    assert got == sorted(want)


def test_opaque_fallback_device():
    vals = np.arange(10_000)
    got = dict(_run_dev(
        Dampr.columns(vals).map(lambda v: v % 13).count()).read())
    want = collections.Counter(int(v) % 13 for v in vals)
    assert got == dict(want)


def test_spill_device():
    rng = np.random.default_rng(5)
    vals = rng.integers(0, 10_000, size=2_000_000)
    got = dict(_run_dev(Dampr.columns(vals).count(),
                        hbm_bytes=1 << 20).read())
    want = collections.Counter(int(v) for v in vals)
    assert got == dict(want)


def test_backend_cross_check():
    """HipOps vs TorchOps on identical inputs."""
    from dampr_amd.gpu.backend import HipOps, TorchOps
    rng = np.random.default_rng(7)
    keys_np = rng.integers(0, 1 << 62, size=300_000, dtype=np.int64)
    vals_np = rng.integers(-100, 100, size=300_000, dtype=np.int64)
    hk = torch.from_numpy(keys_np).cuda()
    hv = torch.from_numpy(vals_np).cuda()
    ck = torch.from_numpy(keys_np)
    cv = torch.from_numpy(vals_np)
    hip, cpu = HipOps(), TorchOps()
    gk, gv = hip.group_reduce(hk, hv, "sum")
    wk, wv = cpu.group_reduce(ck, cv, "sum")
    assert torch.equal(gk.cpu(), wk)
    assert torch.equal(gv.cpu(), wv)
    # partition assignment must agree exactly (CPU tests predict GPU)
    P = 64
    assert torch.equal(hip.partition_of(hk, P).cpu(),
                       cpu.partition_of(ck, P))


def test_mean_device():
    rng = np.random.default_rng(30)
    vals = rng.integers(0, 500, size=500_000)
    got = dict(_run_dev(Dampr.columns(vals).mean(funcs.identity)).read())
    groups = collections.defaultdict(list)
    for v in vals:
        groups[int(v)].append(int(v))
    want = {k: sum(vs) / float(len(vs)) for k, vs in groups.items()}
    assert got == want


def test_topk_device_gpu():
    rng = np.random.default_rng(31)
    vals = rng.integers(-(1 << 50), 1 << 50, size=1_000_000)
    got = _run_dev(Dampr.columns(vals).topk(25)).read()
    want = sorted(sorted((int(v) for v in vals), reverse=True)[:25])
    assert sorted(got) == want


def test_sort_by_device_gpu():
    rng = np.random.default_rng(32)
    vals = rng.integers(-(1 << 40), 1 << 40, size=500_000)
    got = _run_dev(Dampr.columns(vals).sort_by()).read()
    assert got == sorted(int(v) for v in vals)


def test_negative_key_order_gpu():
    vals = np.array([5, -3, 7, -3, 5], dtype=np.int64)
    got = list(_run_dev(Dampr.columns(vals).count()).read())
    assert got == [(-3, 2), (5, 2), (7, 1)]


def test_device_text_df_fused():
    """The reference's own TF-IDF DSL idiom runs on the fused one-pass
    document-frequency kernel."""
    from dampr_amd.gpu.corpus import synth_corpus, oracle_df
    arr = synth_corpus(1 << 20, vocab=5000, seed=21)
    got = dict(_run_dev(Dampr.device_text(arr)
                        .flat_map(funcs.tokenize_set).count()).read())
    want = oracle_df(arr)
    assert got == want


def test_cross_engine_agreement_on_hardware():
    """Host engine (Python operators) vs device engine (HipOps kernels)
    on identical pipelines — the end-to-end numerics oracle on real
    hardware."""
    from dampr_amd.runner import MTRunner
    rng = np.random.default_rng(77)
    vals = rng.integers(-500, 500, size=3000)

    def pipelines(pm):
        return {
            "count": pm.count(),
            "sum": pm.a_group_by().sum(),
            "min": pm.a_group_by().reduce(min),
            "first": pm.a_group_by().first(),
            "topk": pm.topk(9),
            "sort": pm.sort_by(),
            "mean": pm.mean(funcs.identity),
        }

    dev = {name: sorted(p.run(device="cuda:0").read())
           for name, p in pipelines(Dampr.columns(vals)).items()}
    host = {name: sorted(p.run(runner=MTRunner, n_maps=2,
                               n_reducers=2).read())
            for name, p in pipelines(
                Dampr.memory(vals.tolist())).items()}
    for name in dev:
        assert dev[name] == host[name], (name, dev[name][:4],
                                         host[name][:4])


def test_string_keys_device_gpu():
    from dampr_amd.gpu.engine import GpuRunner
    rng = np.random.default_rng(60)
    words = ["w{}".format(int(i))
             for i in rng.integers(0, 5000, size=200_000)]
    got = dict(Dampr.memory(words).count()
               .run(runner=GpuRunner, device="cuda:0").read())
    want = collections.Counter(words)
    assert got == dict(want)


def test_join_probe_batching_device(monkeypatch):
    """Skewed-join guard on HipOps: batched probes + build-side swap
    must match the exact oracle (same pipelines as the CPU-forced
    tests in test_engine.py)."""
    monkeypatch.setenv("DAMPR_JOIN_PROBE_ROWS", "1000")
    rng = np.random.RandomState(7)
    lk = np.concatenate([np.full(5000, 5), rng.randint(0, 50, 500)]) \
        .astype(np.int64)
    lv = rng.randint(1, 100, lk.size).astype(np.int64)
    rk = np.array([5, 5, 7, 9], dtype=np.int64)
    rv = np.array([2, 3, 4, 5], dtype=np.int64)
    got = sorted(_run_dev(
        Dampr.columns(lv, keys=lk).join(Dampr.columns(rv, keys=rk))
        .reduce(funcs.pair_product, many=True)).read())
    rmap = {}
    for j, k in enumerate(rk):
        rmap.setdefault(int(k), []).append(int(rv[j]))
    want = sorted((int(k), int(v) * w)
                  for k, v in zip(lk, lv) for w in rmap.get(int(k), []))
    assert got == want
    # swapped: oversized build side
    got2 = sorted(_run_dev(
        Dampr.columns(rv, keys=rk).join(Dampr.columns(lv, keys=lk))
        .reduce(funcs.pair_product, many=True)).read())
    want2 = sorted((int(k), int(v) * w) for k, v in zip(lk, lv)
                   for w in rmap.get(int(k), []))
    assert got2 == want2


def test_unify_str_stores_device():
    """Cross-encode dictionary remap runs on the HIP backend."""
    from dampr_amd.gpu.engine import DeviceRun, GpuRunner, PartStore
    from dampr_amd.runner import Graph
    dev = torch.device("cuda:0")
    r = GpuRunner("unify-gpu", Graph(), device=dev)
    a = PartStore(str_table=("apple", "fig"))
    a[0] = [DeviceRun(torch.tensor([0, 1, 0], device=dev),
                      torch.tensor([1, 2, 3], device=dev), sorted=True)]
    b = PartStore(str_table=("apple", "kiwi"))
    b[0] = [DeviceRun(torch.tensor([0, 1], device=dev),
                      torch.tensor([4, 5], device=dev), sorted=True)]
    for run in a[0] + b[0]:
        r.pool.admit(run)
    ua, ub = r._unify_str_stores([a, b])
    assert ua.str_table == ("apple", "fig", "kiwi")
    assert sorted(r._decode_store(ua)) == [("apple", 1), ("apple", 3),
                                           ("fig", 2)]
    assert sorted(r._decode_store(ub)) == [("apple", 4), ("kiwi", 5)]
    r.pool.cleanup()


def test_columns_string_keys_join_device():
    """String-keyed columns: dictionary encode at ingest + union remap
    + hash-join kernel on device."""
    lk = np.array(["apple", "apple", "fig", "yam"])
    lv = np.array([1, 2, 3, 4], dtype=np.int64)
    rk = np.array(["apple", "kiwi", "fig"])
    rv = np.array([10, 20, 30], dtype=np.int64)
    out = _run_dev(
        Dampr.columns(lv, keys=lk).join(Dampr.columns(rv, keys=rk))
        .reduce(funcs.pair_sum, many=True))
    assert sorted(out.read()) == [("apple", 11), ("apple", 12),
                                  ("fig", 33)]


# ----------------------------------------------- var-len (string) values

def test_varlen_gather_kernel_oracle():
    """HIP varlen_gather vs a pure-Python reconstruction."""
    from dampr_amd.gpu.strvals import StrVals
    rng = np.random.default_rng(5)
    strings = ["x" * int(n) + str(i)
               for i, n in enumerate(rng.integers(0, 200, size=5000))]
    sv = StrVals.from_strings(strings, device="cuda:0")
    idx_np = rng.integers(0, len(strings), size=8000)
    idx = torch.from_numpy(idx_np).to("cuda:0")
    got = sv.gather(idx).tolist()
    assert got == [strings[i] for i in idx_np]


def test_str_values_join_device():
    lk = np.array([1, 2, 2, 9], dtype=np.int64)
    lv = np.array(["l1", "l2a", "l2b", "l9"])
    rk = np.array([2, 1, 2], dtype=np.int64)
    rv = np.array(["r2a", "r1", "r2b"])
    out = _run_dev(Dampr.columns(lv, keys=lk)
                   .join(Dampr.columns(rv, keys=rk))
                   .reduce(funcs.pair_left, many=True))
    want = [(1, "l1"), (2, "l2a"), (2, "l2a"), (2, "l2b"), (2, "l2b")]
    assert sorted(out.read()) == sorted(want)
    out2 = _run_dev(Dampr.columns(lv, keys=lk)
                    .join(Dampr.columns(rv, keys=rk))
                    .reduce(funcs.pair_right, many=True))
    want2 = [(1, "r1"), (2, "r2a"), (2, "r2b"), (2, "r2a"), (2, "r2b")]
    assert sorted(out2.read()) == sorted(want2)


def test_str_values_large_join_device_cross_backend():
    """Bigger string-value join on HipOps vs the TorchOps CPU result."""
    rng = np.random.default_rng(7)
    lk = rng.integers(0, 5000, size=100_000).astype(np.int64)
    lv = np.array([str(k) + "v" for k in lk])
    rk = rng.integers(0, 5000, size=2000).astype(np.int64)
    rv = np.array(["r" + str(k) for k in rk])

    def build():
        return Dampr.columns(lv, keys=lk) \
            .join(Dampr.columns(rv, keys=rk)) \
            .reduce(funcs.pair_right, many=True)

    gpu = sorted(_run_dev(build()).read())
    cpu = sorted(build().run(device="cpu").read())
    assert gpu == cpu and len(gpu) > 0


def test_str_values_spill_device(tmp_path):
    rng = np.random.default_rng(3)
    lk = rng.integers(0, 64, size=20000).astype(np.int64)
    lv = np.array(["s" * int(i % 33) + str(k)
                   for i, k in enumerate(lk)])
    rk = np.arange(0, 64, 2, dtype=np.int64)
    rv = rk * 3

    def run(**kw):
        return sorted(_run_dev(
            Dampr.columns(lv, keys=lk)
            .join(Dampr.columns(rv, keys=rk))
            .reduce(funcs.pair_left, many=True), **kw).read())

    full = run()
    tiny = run(hbm_bytes=65536, host_bytes=131072,
               spill_dir=str(tmp_path))
    assert tiny == full and len(full) > 0


def test_disk_tier_threaded_device(tmp_path):
    """Force all three tiers on hardware: tiny HBM + host pools push
    runs through the threaded NVMe writes and read-ahead; results must
    match the resident run exactly."""
    rng = np.random.default_rng(21)
    vals = rng.integers(0, 500, size=2_000_000).astype(np.int64)

    def run(**kw):
        return sorted(_run_dev(Dampr.columns(vals).count(),
                               **kw).read())

    full = run()
    tiny = run(hbm_bytes=1 << 21, host_bytes=1 << 21,
               spill_dir=str(tmp_path), n_partitions=32)
    assert tiny == full and len(full) > 0
    # spill files are cleaned up after the job
    left = [f for f in os.listdir(str(tmp_path)) if f.endswith(".run")]
    assert left == [], left


def test_large_join_probe_perm_cross_backend():
    """2M+ probe rows trigger the low-bit probe-clustering permutation;
    emitted indices must map back exactly (TorchOps never permutes, so
    cross-backend equality pins the mapping)."""
    rng = np.random.default_rng(31)
    lk = rng.integers(0, 50_000, size=2_100_000).astype(np.int64)
    lv = rng.integers(0, 1 << 40, size=lk.size).astype(np.int64)
    rk = rng.integers(0, 50_000, size=40_000).astype(np.int64)
    rv = rng.integers(0, 1 << 40, size=rk.size).astype(np.int64)

    def build():
        return Dampr.columns(lv, keys=lk) \
            .join(Dampr.columns(rv, keys=rk)) \
            .reduce(funcs.pair_sum, many=True)

    gpu = sorted(_run_dev(build()).read())
    cpu = sorted(build().run(device="cpu").read())
    assert gpu == cpu and len(gpu) > 0


def test_join_swap_plus_perm_cross_backend(monkeypatch):
    """Build-side swap (skewed right side) combined with the probe
    permutation: cap forced low so the big right side becomes the
    probed side at a size that activates clustering."""
    monkeypatch.setenv("DAMPR_JOIN_PROBE_ROWS", "100000")
    rng = np.random.default_rng(33)
    lk = rng.integers(0, 100_000, size=120_000).astype(np.int64)
    lv = rng.integers(0, 1 << 30, size=lk.size).astype(np.int64)
    rk = rng.integers(0, 100_000, size=1_200_000).astype(np.int64)
    rv = rng.integers(0, 1 << 30, size=rk.size).astype(np.int64)

    def build():
        return Dampr.columns(lv, keys=lk) \
            .join(Dampr.columns(rv, keys=rk)) \
            .reduce(funcs.pair_sum, many=True)

    gpu = sorted(_run_dev(build()).read())
    cpu = sorted(build().run(device="cpu").read())
    assert gpu == cpu and len(gpu) > 0
