// Shared device helpers for the dampr_amd gfx950 kernels.
//
// The hash functions here MUST stay bit-identical to dampr_amd/keyhash.py
// (tests/test_storage.py pins their values) so CPU- and GPU-computed keys
// partition identically.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64          // CDNA4 wavefront width; never 32 here.

typedef unsigned long long u64;
typedef unsigned int u32;
typedef uint8_t u8;

__device__ __forceinline__ u64 fnv1a64_step(u64 h, u8 b) {
    return (h ^ (u64)b) * 0x100000001b3ULL;
}
#define FNV_OFFSET 0xcbf29ce484222325ULL

__device__ __forceinline__ u64 splitmix64(u64 x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

// Order-aware, *parallelizable* token hash (tabulation-rotate): XOR over
// bytes of rotl64(TAB[lower(byte)], pos & 63), finalized with the token
// length.  XOR's associativity lets a wave compute it with a segmented
// scan while a serial loop gets the identical value (mirrored in
// keyhash.tokmix64).  TAB[c] = splitmix64(c + TOKTAB_SEED) — kernels
// materialize it in LDS once per block; serial paths compute entries on
// the fly.  The rotation wraps at 64, so the length finalizer
// disambiguates the (rare) >64-byte tokens' wrapped positions.
#define TOKTAB_SEED 0x7A0BDCAFULL

__device__ __forceinline__ u64 toktab_entry(u8 lowered) {
    return splitmix64((u64)lowered + TOKTAB_SEED);
}

__device__ __forceinline__ u64 rotl64(u64 x, u32 r) {
    r &= 63u;
    return r ? ((x << r) | (x >> (64 - r))) : x;
}

__device__ __forceinline__ u64 tokmix_step(u64 h, u32 pos, u8 lowered) {
    return h ^ rotl64(toktab_entry(lowered), pos);
}

__device__ __forceinline__ u64 tokmix_final(u64 h, u32 len) {
    return h ^ splitmix64((u64)len);
}

// ASCII '\w' classification with lowercasing, matching the reference's
// tokenizer regex r'[^\w]+' (benchmarks/tf-idf-dampr.py:12) on ASCII text.
__device__ __forceinline__ u8 lower_ascii(u8 c) {
    return (c >= 'A' && c <= 'Z') ? (u8)(c + 32) : c;
}
__device__ __forceinline__ bool is_word(u8 c) {
    c = lower_ascii(c);
    return (c >= 'a' && c <= 'z') || (c >= '0' && c <= '9') || c == '_';
}

// Open-addressing u64 table helpers: key 0 is EMPTY (real keys are mixed
// 64-bit hashes; a key hashing to 0 is remapped to 1 by callers).
// Probing reads the key with a plain (cached) load first and only CASes on
// observed-empty: hot keys then cost one read instead of a serializing RMW
// on a contended line.
__device__ __forceinline__ void table_add_u64(u64* __restrict__ keys,
                                              u64* __restrict__ vals,
                                              u64 mask, u64 key, u64 inc) {
    u64 slot = key & mask;
    while (true) {
        u64 cur = keys[slot];
        if (cur == key) { atomicAdd(&vals[slot], inc); return; }
        if (cur == 0ULL) {
            u64 prev = atomicCAS(&keys[slot], 0ULL, key);
            if (prev == 0ULL || prev == key) {
                atomicAdd(&vals[slot], inc);
                return;
            }
        }
        slot = (slot + 1) & mask;
    }
}

// Insert-if-absent; returns true when this call inserted the key.
__device__ __forceinline__ bool table_insert_u64(u64* __restrict__ keys,
                                                 u64 mask, u64 key,
                                                 u64* slot_out) {
    u64 slot = key & mask;
    while (true) {
        u64 cur = keys[slot];
        if (cur == key) { *slot_out = slot; return false; }
        if (cur == 0ULL) {
            u64 prev = atomicCAS(&keys[slot], 0ULL, key);
            if (prev == 0ULL) { *slot_out = slot; return true; }
            if (prev == key)  { *slot_out = slot; return false; }
        }
        slot = (slot + 1) & mask;
    }
}
