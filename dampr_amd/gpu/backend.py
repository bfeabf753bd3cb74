"""Device-op backends for the columnar engine.

Two implementations of one small interface (sort, segmented reduce, hash
join, partition histogram):

* ``HipOps`` — the hand-written gfx950 kernels (ops/hip/*.hip) via the
  dampr_hip extension.  **The only backend allowed on a CUDA device**: if
  the extension is missing on a GPU box this raises instead of silently
  falling back to eager PyTorch.
* ``TorchOps`` — a plain-PyTorch oracle used (a) as the CPU reference the
  HIP kernels are numerics-tested against and (b) to run the full engine
  logic in the no-GPU CI (tests/, gloo world>1).

Pick with ``ops_for(device)``.
"""
import torch

OP_SUM, OP_MIN, OP_MAX = 0, 1, 2
_OP_IDS = {"sum": OP_SUM, "min": OP_MIN, "max": OP_MAX}

_I64_MIN = -(1 << 63)
_I64_MAX = (1 << 63) - 1


def ops_for(device):
    device = torch.device(device)
    if device.type == "cuda":
        return HipOps()
    return TorchOps()


class _OpsBase(object):
    def sort_pairs(self, keys, payload=None):
        raise NotImplementedError

    def seg_reduce_sorted(self, sorted_keys, vals, op="sum"):
        raise NotImplementedError

    def hash_join(self, keys_l, keys_r, how="inner", table=None):
        raise NotImplementedError

    def hash_join_build(self, keys_r):
        """Opaque reusable build-side state for batched probing."""
        return None

    def probe_order(self, keys):
        """Optional probe-locality permutation (None = keep order)."""
        return None

    def merge_sorted_runs(self, ks, fkeys=False):
        """K-way merge of key-sorted runs (K4).  ``ks`` is a list of
        runs each already in host-ascending order (signed for i64 keys,
        encoded-unsigned for fkeys).  Returns (merged_keys, perm) where
        perm indexes into the runs' concatenation.  Stable: ties keep
        run order, then within-run order."""
        raise NotImplementedError

    # shared helpers ------------------------------------------------------

    def partition_of(self, keys, n_partitions):
        """Partition id per key: hash-mix then mod (K1).  Must agree across
        backends so CPU tests predict device placement."""
        # splitmix64-style finalizer over the i64 bit pattern
        x = keys.to(torch.int64)
        x = x ^ (x >> 30)
        x = x * -4658895280553007687          # 0xbf58476d1ce4e5b9 as i64
        x = x ^ (x >> 27)
        x = x * -7723592293110705685          # 0x94d049bb133111eb as i64
        x = x ^ (x >> 31)
        return torch.remainder(x, n_partitions)

    def group_reduce(self, keys, vals, op="sum"):
        """Unsorted group-by reduce: sort then segmented reduce."""
        sk, sp = self.sort_pairs(keys)
        sv = vals[sp.to(torch.int64)]
        return self.seg_reduce_sorted(sk, sv, op)

    @staticmethod
    def segment_ids(sorted_keys):
        n = sorted_keys.numel()
        if n == 0:
            z = torch.zeros(0, dtype=torch.int64, device=sorted_keys.device)
            return z, sorted_keys
        flags = torch.ones(n, dtype=torch.int64, device=sorted_keys.device)
        flags[1:] = (sorted_keys[1:] != sorted_keys[:-1]).to(torch.int64)
        seg = torch.cumsum(flags, 0) - 1
        uniq = sorted_keys[flags.bool()]
        return seg, uniq


class HipOps(_OpsBase):
    """gfx950 HIP kernel backend (see gpu/relational.py docstrings for the
    kernel-level notes)."""

    def __init__(self):
        from ..ops import native
        self.ext = native.require()

    def sort_pairs(self, keys, payload=None):
        from .relational import radix_sort_pairs
        return radix_sort_pairs(keys, payload)

    def seg_reduce_sorted(self, sorted_keys, vals, op="sum"):
        from .relational import group_reduce_sorted
        return group_reduce_sorted(sorted_keys, vals, _OP_IDS[op])

    def hash_join(self, keys_l, keys_r, how="inner", table=None):
        from .relational import hash_join
        return hash_join(keys_l, keys_r, how, table=table)

    def hash_join_build(self, keys_r):
        from .relational import hj_build_table
        return hj_build_table(keys_r)

    def probe_order(self, keys):
        """Permutation clustering probe keys by their low bits (hash
        slot = key & mask): probes then walk L2-resident table regions
        instead of random HBM lines.  Two low-byte passes give 64K-key
        clusters (~1 MB table window each)."""
        if keys.numel() < (1 << 20):
            return None
        from .relational import radix_sort_pairs
        _k, perm = radix_sort_pairs(keys, low_passes=2)
        return perm.to(torch.int64)

    def merge_sorted_runs(self, ks, fkeys=False):
        total = sum(k.numel() for k in ks)
        if total >= (1 << 31):      # u32 payload limit (same as sort)
            raise OverflowError("merge payload exceeds u32")
        dev = ks[0].device
        items = []
        base = 0
        for k in ks:
            p = torch.arange(base, base + k.numel(), dtype=torch.int32,
                             device=dev)
            items.append((k.contiguous(), p))
            base += k.numel()
        bias = 0 if fkeys else 1    # signed compare for raw i64 keys
        # pairwise merge tree over ADJACENT runs (keeps global stability)
        while len(items) > 1:
            nxt = []
            for a in range(0, len(items) - 1, 2):
                ka, pa = items[a]
                kb, pb = items[a + 1]
                n = ka.numel() + kb.numel()
                out_k = torch.empty(n, dtype=ka.dtype, device=dev)
                out_p = torch.empty(n, dtype=torch.int32, device=dev)
                self.ext.mp_merge(ka, pa, kb, pb, bias, out_k, out_p)
                nxt.append((out_k, out_p))
            if len(items) % 2:
                nxt.append(items[-1])
            items = nxt
        k, p = items[0]
        return k, p.to(torch.int64)


class TorchOps(_OpsBase):
    """Pure-torch oracle (CPU tests; never used on a CUDA device)."""

    def sort_pairs(self, keys, payload=None):
        n = keys.numel()
        if payload is None:
            payload = torch.arange(n, dtype=torch.int32, device=keys.device)
        if n <= 1:
            return keys.clone(), payload.clone()
        # unsigned order on the i64 bit pattern = flip sign bit, signed sort
        flipped = keys ^ _I64_MIN
        order = torch.argsort(flipped, stable=True)
        return keys[order], payload[order]

    def seg_reduce_sorted(self, sorted_keys, vals, op="sum"):
        seg, uniq = self.segment_ids(sorted_keys)
        n_seg = uniq.numel()
        if vals.dtype == torch.float64:
            init = {"sum": 0.0, "min": float("inf"),
                    "max": float("-inf")}[op]
        else:
            init = {"sum": 0, "min": _I64_MAX, "max": _I64_MIN}[op]
        out = torch.full((max(n_seg, 1),), init, dtype=vals.dtype,
                         device=vals.device)
        red = {"sum": "sum", "min": "amin", "max": "amax"}[op]
        out[:n_seg] = out[:n_seg].scatter_reduce(
            0, seg, vals, reduce=red, include_self=True)
        return uniq, out[:n_seg]

    def merge_sorted_runs(self, ks, fkeys=False):
        keys = torch.cat(ks)
        ordk = (keys ^ _I64_MIN) if fkeys else keys
        perm = torch.argsort(ordk, stable=True)
        return keys[perm], perm

    def hash_join_build(self, keys_r):
        return self.sort_pairs(keys_r)

    def hash_join(self, keys_l, keys_r, how="inner", table=None):
        assert how in ("inner", "left", "outer")
        dev = keys_l.device
        # sort right side; binary-search each left key's run
        sr, pr = table if table is not None else self.sort_pairs(keys_r)
        fl = keys_l ^ _I64_MIN
        fr = sr ^ _I64_MIN
        lo = torch.searchsorted(fr, fl, side="left")
        hi = torch.searchsorted(fr, fl, side="right")
        counts = hi - lo
        matched_l = counts > 0
        if how in ("left", "outer"):
            counts = torch.where(matched_l, counts,
                                 torch.ones_like(counts))
        else:
            counts = torch.where(matched_l, counts,
                                 torch.zeros_like(counts))
        offs = torch.cumsum(counts, 0) - counts
        total = int(counts.sum().item())
        out_l = torch.empty(total, dtype=torch.int64, device=dev)
        out_r = torch.empty(total, dtype=torch.int64, device=dev)
        pos = torch.arange(total, device=dev)
        src = torch.repeat_interleave(
            torch.arange(keys_l.numel(), device=dev), counts)
        out_l[pos] = src
        within = pos - offs[src]
        has = matched_l[src]
        rr = torch.where(
            has, pr.to(torch.int64)[
                torch.clamp(lo[src] + within, max=max(sr.numel() - 1, 0))],
            torch.full_like(src, -1))
        out_r[pos] = rr
        if how == "outer" and keys_r.numel():
            m = torch.zeros(keys_r.numel(), dtype=torch.bool, device=dev)
            mr = out_r[out_r >= 0]
            m[mr] = True
            un = torch.nonzero(~m).flatten()
            if un.numel():
                out_l = torch.cat([out_l, torch.full_like(un, -1)])
                out_r = torch.cat([out_r, un])
        return out_l, out_r
