"""Stable 64-bit key hashing, shared between the CPU engine and the HIP kernels.

The reference partitions records with Python's salted ``hash(key) % n``
(reference: base.py:6-8), which is only stable within one fork tree.  The
MI355X engine needs a hash that the device kernels can reproduce bit-for-bit
(K1 in SURVEY.md §2.4), so keys of the common types (int / str / bytes) are
hashed with splitmix64 / FNV-1a-64 — the exact functions implemented in
``ops/hip/common.h``.  Other Python objects fall back to the interpreter hash
(CPU-only path; stable within a run because workers are forked).
"""

_MASK = (1 << 64) - 1

FNV_OFFSET = 0xcbf29ce484222325
FNV_PRIME = 0x100000001b3


def fnv1a64(data: bytes) -> int:
    """FNV-1a over bytes; must match fnv1a64() in ops/hip/common.h."""
    h = FNV_OFFSET
    for b in data:
        h = ((h ^ b) * FNV_PRIME) & _MASK
    return h


def splitmix64(x: int) -> int:
    """splitmix64 finalizer; must match splitmix64() in ops/hip/common.h."""
    x = (x + 0x9E3779B97F4A7C15) & _MASK
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & _MASK
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & _MASK
    return x ^ (x >> 31)


_TOKTAB_SEED = 0x7A0BDCAF


def tokmix64(data: bytes) -> int:
    """Order-aware tabulation-rotate token hash; must match
    tokmix_step()/tokmix_final() in ops/hip/common.h (the GPU computes it
    with a wave-wide segmented XOR scan; XOR's associativity makes serial
    and parallel answers equal).  Bytes are lowercased like the device
    tokenizer; the length finalizer disambiguates >64-byte rotation
    wraps."""
    h = 0
    for j, b in enumerate(data):
        c = b + 32 if 65 <= b <= 90 else b
        t = splitmix64(c + _TOKTAB_SEED)
        r = j & 63
        h ^= ((t << r) | (t >> (64 - r))) & _MASK if r else t
    return h ^ splitmix64(len(data))


def key_hash64(key) -> int:
    """Canonical u64 hash of a record key."""
    if isinstance(key, bool):          # bool before int: True is an int
        return splitmix64(int(key))
    if isinstance(key, int):
        return splitmix64(key & _MASK)
    if isinstance(key, str):
        return fnv1a64(key.encode("utf-8"))
    if isinstance(key, bytes):
        return fnv1a64(key)
    # Arbitrary object: interpreter hash (stable across forked workers).
    return hash(key) & _MASK


def partition_of(key, n_partitions: int) -> int:
    """Partition assignment; device kernels use hash % n identically."""
    return key_hash64(key) % n_partitions


def stable_hash64(key) -> int:
    """key_hash64, but process-independent for EVERY key type: arbitrary
    objects hash via their repr instead of the per-process-salted
    interpreter hash.  Used for cross-rank ownership decisions, where
    all ranks must compute identical values (spawned ranks do not share
    a hash salt the way forked workers do)."""
    if isinstance(key, (bool, int, str, bytes)):
        return key_hash64(key)
    return fnv1a64(repr(key).encode("utf-8"))
