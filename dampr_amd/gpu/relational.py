"""Device relational ops over (u64-key, payload) columns: sort, group-by
reduce, join, top-k.

These are the GPU engine's building blocks for the DSL's ``group_by``,
``sort_by``, ``join`` and ``topk`` when keys/values lower to device columns
(SURVEY.md §2.4 K2/K3/K5/K7/K8/K11).  Keys are 64-bit (keyhash.key_hash64
for strings); payloads are row indices into host or device value arenas.

torch is the tensor layer (allocations, cumsum/nonzero metadata glue); all
per-record hot work is in the hand-written HIP kernels.
"""
import torch

from ..ops import native

OP_SUM, OP_MIN, OP_MAX = 0, 1, 2

_I64_MIN = -(1 << 63)
_I64_MAX = (1 << 63) - 1


def _pow2_at_least(n):
    return 1 << max(4, int(n - 1).bit_length())


def radix_sort_pairs(keys, payload=None, low_passes=None):
    """Stable LSD radix sort of int64-as-u64 keys; returns (keys, payload)
    sorted in unsigned key order.  Skips passes whose digit is constant.
    ``low_passes`` stops after that many ACTIVE low-byte passes — a
    partial sort that clusters keys by their low bits (hash-table probe
    locality wants slot-region grouping, not total order)."""
    ext = native.require()
    n = keys.numel()
    if payload is None:
        payload = torch.arange(n, dtype=torch.int32, device=keys.device)
    if n <= 1:
        return keys.clone(), payload.clone()
    RS_SPAN = ext.rs_span()
    nblocks = (n + RS_SPAN - 1) // RS_SPAN
    # The caller's tensors are read-only: the first executed pass scatters
    # out of them into an owned buffer, and later passes ping-pong between
    # two owned buffers (using the input as scratch would clobber it).
    src_k, src_p = keys.contiguous(), payload.contiguous()
    out_k, out_p = torch.empty_like(src_k), torch.empty_like(src_p)
    # one streaming read decides which byte passes are constant
    # (skippable): digit d is constant iff its byte of the AND-fold
    # equals its byte of the OR-fold.  Single host sync for the whole
    # sort; runs at read bandwidth (register fold + one atomic/wave).
    a, o = ext.rs_digit_fold(src_k).cpu().tolist()
    active = [((a >> (8 * b)) & 255) != ((o >> (8 * b)) & 255)
              for b in range(8)]
    n_done = 0
    for byte in range(8):
        if low_passes is not None and n_done >= low_passes:
            break
        shift = byte * 8
        if not active[byte]:
            continue                      # constant digit: skip pass
        hist = ext.rs_hist(src_k, shift, nblocks)
        scanned = torch.cumsum(hist, 0, dtype=torch.int64)
        scanned -= hist                # exclusive, fused i64 upcast
        ext.rs_scatter(src_k, src_p, scanned, shift, nblocks, out_k, out_p)
        n_done += 1
        if n_done == 1:
            src_k, src_p = out_k, out_p
            out_k, out_p = (torch.empty_like(src_k),
                            torch.empty_like(src_p))
        else:
            src_k, src_p, out_k, out_p = out_k, out_p, src_k, src_p
    if n_done == 0:
        return src_k.clone(), src_p.clone()
    return src_k, src_p


def segment_ids(sorted_keys):
    """(seg_ids i64, unique_keys) for a key-sorted column."""
    n = sorted_keys.numel()
    if n == 0:
        return (torch.zeros(0, dtype=torch.int64,
                            device=sorted_keys.device), sorted_keys)
    flags = torch.ones(n, dtype=torch.int64, device=sorted_keys.device)
    flags[1:] = (sorted_keys[1:] != sorted_keys[:-1]).to(torch.int64)
    seg = torch.cumsum(flags, 0) - 1
    uniq = sorted_keys[flags.bool()]
    return seg, uniq


def group_reduce_sorted(sorted_keys, vals, op=OP_SUM):
    """Segmented reduce over a key-sorted column; returns (unique_keys,
    aggregates).  Fused form (K5+K7): per-tile segment-start counts +
    a tiny tile-base scan, then ONE pass that re-derives boundaries
    from neighbor keys, emits each segment head's key and folds values
    — no materialized per-element segment ids (the flags -> cumsum ->
    nonzero -> gather chain cost 4+ full-column passes and a host
    sync)."""
    ext = native.require()
    n = sorted_keys.numel()
    if n == 0:
        return sorted_keys, vals[:0]
    sorted_keys = sorted_keys.contiguous()
    vals = vals.contiguous()
    counts = ext.seg_count(sorted_keys)
    scan = torch.cumsum(counts, 0, dtype=torch.int64)
    tile_base = scan - counts
    n_seg = int(scan[-1].item())
    if vals.dtype == torch.float64:
        init = {OP_SUM: 0.0, OP_MIN: float("inf"),
                OP_MAX: float("-inf")}[op]
        out = torch.full((n_seg,), init, dtype=torch.float64,
                         device=vals.device)
    else:
        init = {OP_SUM: 0, OP_MIN: _I64_MAX, OP_MAX: _I64_MIN}[op]
        out = torch.full((n_seg,), init, dtype=torch.int64,
                         device=vals.device)
    uniq = torch.empty(n_seg, dtype=torch.int64,
                       device=sorted_keys.device)
    ext.seg_reduce_fused(sorted_keys, vals, tile_base, op, uniq, out)
    return uniq, out


def group_sum(keys, vals):
    """Unsorted group-by-sum: radix sort then segmented reduce (the
    high-cardinality path; for low-cardinality the hash-combine table in
    tfidf.py is the faster route)."""
    sk, sp = radix_sort_pairs(keys)
    sv = vals[sp.to(torch.int64)]
    return group_reduce_sorted(sk, sv)


def hj_build_table(keys_r):
    """Build the chained hash table over the right side once; reusable
    across probe batches (the skew guard probes in chunks — rebuilding
    the table per chunk cost ~1.6 ms x batches at 25M rows)."""
    ext = native.require()
    dev = keys_r.device
    nr = keys_r.numel()
    cap = _pow2_at_least(max(2 * nr, 16))
    t_keys = torch.zeros(cap, dtype=torch.int64, device=dev)
    # cap+1 heads: t_head[cap] is the dedicated zero-key chain (raw keys
    # include 0 — dictionary rank ids, int columns — and the open table
    # uses 0 as EMPTY; see hj_build_kernel)
    t_head = torch.full((cap + 1,), -1, dtype=torch.int64, device=dev)
    nxt = torch.empty(max(nr, 1), dtype=torch.int64, device=dev)
    ext.hj_build(keys_r, t_keys, t_head, nxt)
    return (t_keys, t_head, nxt, nr)


def hash_join(keys_l, keys_r, how="inner", table=None):
    """Equi-join on u64 keys.  Returns (l_idx, r_idx) int64 row-index pairs;
    for "left"/"outer", missing matches carry index -1.

    how: "inner" | "left" | "outer"; ``table`` reuses hj_build_table.
    """
    assert how in ("inner", "left", "outer")
    ext = native.require()
    t_keys, t_head, nxt, nr = table or hj_build_table(keys_r)
    left_outer = 1 if how in ("left", "outer") else 0
    counts = ext.hj_count(keys_l, t_keys, t_head, nxt, left_outer)
    offsets = torch.cumsum(counts, 0) - counts
    total = int(counts.sum().item())
    out_l, out_r, matched = ext.hj_emit(
        keys_l, t_keys, t_head, nxt, offsets, total, left_outer,
        1 if how == "outer" else 0, nr)
    out_l, out_r = out_l[:total], out_r[:total]
    if how == "outer" and nr:
        un = torch.nonzero(matched[:nr] == 0).flatten()
        if un.numel():
            out_l = torch.cat([out_l, torch.full_like(un, -1)])
            out_r = torch.cat([out_r, un])
    return out_l, out_r


def topk_by(values_u64, k, largest=True):
    """Top-k row indices ordered by a u64 sort column (K11 via the radix
    sort; the column is a key-encoded ordering, e.g. flipped-sign floats)."""
    keys = values_u64 if not largest else ~values_u64
    sk, sp = radix_sort_pairs(keys)
    return sp[:k].to(torch.int64)


def encode_f64_sortable(x):
    """Map float64 to u64 preserving order (IEEE trick): sortable keys for
    sort_by/topk on floats.  ``+ 0.0`` canonicalizes -0.0 to +0.0 first
    (Python == merges them; distinct bit patterns would split the
    group), and leaves every other value including NaN unchanged."""
    b = (x + 0.0).view(torch.int64)
    neg = b < 0
    sign_bit = -(1 << 63)                 # 0x8000000000000000 as int64
    out = torch.where(neg, ~b, b ^ sign_bit)
    return out
