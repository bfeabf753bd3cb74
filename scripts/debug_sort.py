"""Debug harness for the radix sort payload mismatch seen on GPU."""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from dampr_amd.ops import native

DEV = "cuda:0"
RS_SPAN = 4096


def one_pass(keys, payload, shift):
    ext = native.require()
    n = keys.numel()
    nblocks = (n + RS_SPAN - 1) // RS_SPAN
    hist = ext.rs_hist(keys, shift, nblocks).to(torch.int64)
    scanned = torch.cumsum(hist, 0) - hist
    out_k = torch.empty_like(keys)
    out_p = torch.empty_like(payload)
    ext.rs_scatter(keys, payload, scanned, shift, nblocks, out_k, out_p)
    torch.cuda.synchronize()
    return out_k, out_p


def check_pass(tag, keys, payload, shift):
    out_k, out_p = one_pass(keys, payload, shift)
    n = keys.numel()
    ok_perm = torch.equal(torch.sort(out_p.to(torch.int64)).values,
                          torch.arange(n, device=keys.device))
    ok_pair = torch.equal(keys[out_p.to(torch.int64)], out_k)
    km = torch.equal(torch.sort(out_k).values, torch.sort(keys).values)
    print(f"{tag}: n={n} perm_ok={ok_perm} pair_ok={ok_pair} "
          f"multiset_ok={km}")
    if not (ok_perm and ok_pair):
        bad = (keys[out_p.to(torch.int64)] != out_k).nonzero().flatten()
        print("  first bad idx:", bad[:10].tolist())
        if bad.numel():
            i = int(bad[0])
            print("  at", i, "out_k", hex(int(out_k[i])), "keys[out_p]",
                  hex(int(keys[out_p[i]])), "out_p", int(out_p[i]))
    return ok_perm and ok_pair


def main():
    rng = np.random.default_rng(0)
    ext = native.require()  # noqa

    # tiny: 1 block, 1 tile
    k = torch.from_numpy(
        rng.integers(0, 1 << 63, size=200, dtype=np.int64)).to(DEV)
    p = torch.arange(200, dtype=torch.int32, device=DEV)
    check_pass("tiny-200", k, p, 0)

    # 1 block, multi-tile
    k = torch.from_numpy(
        rng.integers(0, 1 << 63, size=3000, dtype=np.int64)).to(DEV)
    p = torch.arange(3000, dtype=torch.int32, device=DEV)
    check_pass("one-block-3000", k, p, 0)

    # multi-block
    for n in (5000, 100_000, 1_000_000):
        k = torch.from_numpy(
            rng.integers(0, 1 << 63, size=n, dtype=np.int64)).to(DEV)
        p = torch.arange(n, dtype=torch.int32, device=DEV)
        ok = True
        for shift in (0, 8, 16):
            ok = check_pass(f"multi-{n}-s{shift}", k, p, shift) and ok

    # full sort small + big
    from dampr_amd.gpu.relational import radix_sort_pairs
    for n in (3000, 1_000_000):
        kn = rng.integers(0, 1 << 63, size=n, dtype=np.int64)
        k = torch.from_numpy(kn).to(DEV)
        sk, sp = radix_sort_pairs(k)
        torch.cuda.synchronize()
        ok_keys = np.array_equal(sk.cpu().numpy().view(np.uint64),
                                 np.sort(kn.view(np.uint64)))
        ok_pair = torch.equal(k[sp.to(torch.int64)], sk)
        ok_perm = torch.equal(torch.sort(sp.to(torch.int64)).values,
                              torch.arange(n, device=DEV))
        print(f"fullsort-{n}: keys_ok={ok_keys} pair_ok={ok_pair} "
              f"perm_ok={ok_perm}")


if __name__ == "__main__":
    main()
