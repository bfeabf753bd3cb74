"""GPU relational ops (sort / group / join / topk) vs exact CPU oracles."""
import numpy as np
import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif("not __import__('torch').cuda.is_available()",
                       reason="needs an MI355X"),
]

if torch.cuda.is_available():
    DEV = torch.device("cuda:0")
else:
    DEV = None


@pytest.fixture(scope="module")
def rng():
    return np.random.default_rng(7)


def test_radix_sort_random(rng):
    from dampr_amd.gpu.relational import radix_sort_pairs
    n = 1_000_000
    keys_np = rng.integers(0, 1 << 63, size=n, dtype=np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    sk, sp = radix_sort_pairs(keys)
    want = np.sort(keys_np.view(np.uint64))
    np.testing.assert_array_equal(sk.cpu().numpy().view(np.uint64), want)
    # payload is the permutation
    assert torch.equal(keys[sp.to(torch.int64)], sk)


def test_radix_sort_stability(rng):
    from dampr_amd.gpu.relational import radix_sort_pairs
    # few distinct keys: payload order within a key must be preserved
    keys_np = rng.integers(0, 8, size=100_000, dtype=np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    sk, sp = radix_sort_pairs(keys)
    sp_np = sp.cpu().numpy().astype(np.int64)
    sk_np = sk.cpu().numpy()
    for key in range(8):
        mask = sk_np == key
        assert (np.diff(sp_np[mask]) > 0).all()


def test_radix_sort_negative_as_unsigned(rng):
    from dampr_amd.gpu.relational import radix_sort_pairs
    keys_np = rng.integers(-(1 << 62), 1 << 62, size=50_000,
                           dtype=np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    sk, _ = radix_sort_pairs(keys)
    want = np.sort(keys_np.view(np.uint64))
    np.testing.assert_array_equal(sk.cpu().numpy().view(np.uint64), want)


def test_group_sum_vs_numpy(rng):
    from dampr_amd.gpu.relational import group_sum
    n = 500_000
    keys_np = rng.zipf(1.5, size=n).astype(np.int64) % 10_000
    vals_np = rng.integers(-100, 100, size=n).astype(np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    vals = torch.from_numpy(vals_np).to(DEV)
    uk, agg = group_sum(keys, vals)
    got = dict(zip(uk.cpu().numpy().tolist(), agg.cpu().numpy().tolist()))
    want = {}
    for k, v in zip(keys_np.tolist(), vals_np.tolist()):
        want[k] = want.get(k, 0) + v
    assert got == want


def test_group_minmax(rng):
    from dampr_amd.gpu.relational import (group_reduce_sorted,
                                          radix_sort_pairs, OP_MIN, OP_MAX)
    keys_np = rng.integers(0, 50, size=20_000, dtype=np.int64)
    vals_np = rng.integers(-1000, 1000, size=20_000, dtype=np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    vals = torch.from_numpy(vals_np).to(DEV)
    sk, sp = radix_sort_pairs(keys)
    sv = vals[sp.to(torch.int64)]
    uk, mn = group_reduce_sorted(sk, sv, OP_MIN)
    _, mx = group_reduce_sorted(sk, sv, OP_MAX)
    import collections
    want_mn = collections.defaultdict(lambda: 10**9)
    want_mx = collections.defaultdict(lambda: -10**9)
    for k, v in zip(keys_np.tolist(), vals_np.tolist()):
        want_mn[k] = min(want_mn[k], v)
        want_mx[k] = max(want_mx[k], v)
    uk_np = uk.cpu().numpy().tolist()
    assert dict(zip(uk_np, mn.cpu().numpy().tolist())) == dict(want_mn)
    assert dict(zip(uk_np, mx.cpu().numpy().tolist())) == dict(want_mx)


def test_group_sum_f64(rng):
    from dampr_amd.gpu.relational import (group_reduce_sorted,
                                          radix_sort_pairs)
    keys_np = rng.integers(0, 100, size=50_000, dtype=np.int64)
    vals_np = rng.standard_normal(50_000)
    keys = torch.from_numpy(keys_np).to(DEV)
    vals = torch.from_numpy(vals_np).to(DEV)
    sk, sp = radix_sort_pairs(keys)
    uk, s = group_reduce_sorted(sk, vals[sp.to(torch.int64)])
    want = np.zeros(100)
    np.add.at(want, keys_np, vals_np)
    got = np.zeros(100)
    got[uk.cpu().numpy()] = s.cpu().numpy()
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-9)


def _join_oracle(lk, rk, how):
    out = []
    from collections import defaultdict
    rmap = defaultdict(list)
    for j, k in enumerate(rk):
        rmap[k].append(j)
    matched_r = set()
    for i, k in enumerate(lk):
        if rmap[k]:
            for j in rmap[k]:
                out.append((i, j))
                matched_r.add(j)
        elif how in ("left", "outer"):
            out.append((i, -1))
    if how == "outer":
        for j in range(len(rk)):
            if j not in matched_r:
                out.append((-1, j))
    return sorted(out)


@pytest.mark.parametrize("how", ["inner", "left", "outer"])
def test_hash_join(rng, how):
    from dampr_amd.gpu.relational import hash_join
    lk_np = rng.integers(1, 500, size=3000, dtype=np.int64)
    rk_np = rng.integers(250, 750, size=2000, dtype=np.int64)
    li, ri = hash_join(torch.from_numpy(lk_np).to(DEV),
                       torch.from_numpy(rk_np).to(DEV), how=how)
    got = sorted(zip(li.cpu().numpy().tolist(),
                     ri.cpu().numpy().tolist()))
    assert got == _join_oracle(lk_np.tolist(), rk_np.tolist(), how)


@pytest.mark.parametrize("how", ["inner", "left", "outer"])
def test_hash_join_zero_and_one_keys(how):
    """Regression (round-1 driver failure): the open table uses key 0 as
    its EMPTY sentinel and the old remap ``if (!k) k = 1`` aliased key 0
    with key 1 — dictionary rank ids start at 0, so "apple"(0) joined
    "fig"(1)'s rows.  Keys 0 and 1 must join independently."""
    from dampr_amd.gpu.relational import hash_join
    lk_np = np.array([0, 1, 0, 2, 1, 0], dtype=np.int64)
    rk_np = np.array([1, 0, 3, 0], dtype=np.int64)
    li, ri = hash_join(torch.from_numpy(lk_np).to(DEV),
                       torch.from_numpy(rk_np).to(DEV), how=how)
    got = sorted(zip(li.cpu().numpy().tolist(),
                     ri.cpu().numpy().tolist()))
    assert got == _join_oracle(lk_np.tolist(), rk_np.tolist(), how)


@pytest.mark.parametrize("how", ["inner", "left", "outer"])
def test_hash_join_i64_extremes(how):
    from dampr_amd.gpu.relational import hash_join
    ext = [0, 1, -1, (1 << 63) - 1, -(1 << 63), 2, -2]
    lk_np = np.array(ext + [0, (1 << 63) - 1], dtype=np.int64)
    rk_np = np.array([-(1 << 63), 0, 5, (1 << 63) - 1, -1],
                     dtype=np.int64)
    li, ri = hash_join(torch.from_numpy(lk_np).to(DEV),
                       torch.from_numpy(rk_np).to(DEV), how=how)
    got = sorted(zip(li.cpu().numpy().tolist(),
                     ri.cpu().numpy().tolist()))
    assert got == _join_oracle(lk_np.tolist(), rk_np.tolist(), how)


def test_hash_join_zero_key_heavy(rng):
    """Many zero keys (dedicated chain must carry real multiplicity)."""
    from dampr_amd.gpu.relational import hash_join
    lk_np = rng.integers(0, 4, size=5000, dtype=np.int64)
    rk_np = rng.integers(0, 4, size=300, dtype=np.int64)
    li, ri = hash_join(torch.from_numpy(lk_np).to(DEV),
                       torch.from_numpy(rk_np).to(DEV), how="inner")
    got = sorted(zip(li.cpu().numpy().tolist(),
                     ri.cpu().numpy().tolist()))
    assert got == _join_oracle(lk_np.tolist(), rk_np.tolist(), "inner")


def test_topk(rng):
    from dampr_amd.gpu.relational import topk_by, encode_f64_sortable
    vals_np = rng.standard_normal(100_000)
    vals = torch.from_numpy(vals_np).to(DEV)
    enc = encode_f64_sortable(vals)
    idx = topk_by(enc, 10, largest=True).cpu().numpy()
    want = np.argsort(-vals_np)[:10]
    np.testing.assert_array_equal(np.sort(vals_np[idx])[::-1],
                                  np.sort(vals_np[want])[::-1])


def test_radix_sort_input_not_clobbered(rng):
    """Regression: multi-pass sort once used the caller's tensors as
    ping-pong scratch, overwriting the input after two passes."""
    from dampr_amd.gpu.relational import radix_sort_pairs
    n = 100_000
    keys_np = rng.integers(0, 1 << 63, size=n, dtype=np.int64)
    keys = torch.from_numpy(keys_np).to(DEV)
    sk, sp = radix_sort_pairs(keys)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(keys.cpu().numpy(), keys_np)
    assert torch.equal(keys[sp.to(torch.int64)], sk)


def test_merge_sorted_runs_oracle(rng):
    """Merge-path k-way merge vs numpy: signed i64 order, ties stable
    (run order, then within-run order)."""
    from dampr_amd.gpu.backend import HipOps
    ops = HipOps()
    runs_np = []
    for r in range(7):                       # odd count: tree leftover
        n = int(rng.integers(1, 40_000))
        a = np.sort(rng.integers(-1000, 1000, size=n, dtype=np.int64))
        runs_np.append(a)
    ks = [torch.from_numpy(a).to(DEV) for a in runs_np]
    mk, perm = ops.merge_sorted_runs(ks, fkeys=False)
    cat = np.concatenate(runs_np)
    order = np.argsort(cat, kind="stable")
    np.testing.assert_array_equal(mk.cpu().numpy(), cat[order])
    np.testing.assert_array_equal(perm.cpu().numpy(), order)


def test_merge_sorted_runs_unsigned_fkeys(rng):
    from dampr_amd.gpu.backend import HipOps
    ops = HipOps()
    runs_np = []
    for r in range(3):
        a = rng.integers(0, 1 << 63, size=30_000, dtype=np.int64) \
            .view(np.uint64)
        a = np.sort(a).view(np.int64)        # unsigned ascending
        runs_np.append(a)
    ks = [torch.from_numpy(a).to(DEV) for a in runs_np]
    mk, perm = ops.merge_sorted_runs(ks, fkeys=True)
    cat = np.concatenate(runs_np)
    order = np.argsort(cat.view(np.uint64), kind="stable")
    np.testing.assert_array_equal(mk.cpu().numpy(), cat[order])


def test_merge_sorted_runs_empty_and_single():
    from dampr_amd.gpu.backend import HipOps
    ops = HipOps()
    e = torch.zeros(0, dtype=torch.int64, device=DEV)
    a = torch.tensor([1, 2, 3], dtype=torch.int64, device=DEV)
    mk, perm = ops.merge_sorted_runs([e, a, e], fkeys=False)
    assert mk.cpu().tolist() == [1, 2, 3]
    assert perm.cpu().tolist() == [0, 1, 2]
