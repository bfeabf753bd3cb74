// dampr_amd gfx950 kernels: text ingest, tokenize+hash (K1), device
// hash-table combine (K6), extraction, broadcast-apply (K9) and the idf
// epilogue.  These are the MI355X-native replacements for the reference's
// per-record Python hot loops; the roles, not the code,
// come from dampr/dataset.py + dampr/base.py.
//
// Design notes (cdna_hip_programming.md):
//  - memory-bound kernels: 256-thread blocks, vectorized 16 B/lane loads,
//    grid capped with grid-stride loops (Guideline 11/13).
//  - wave64 ballots (u64 masks) for in-block stable compaction.
//  - all inter-block communication via device-scope atomics on HBM.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"

#define BLOCK 256
#define VBYTES 16                     // bytes per lane per iteration
#define TILE (BLOCK * VBYTES)         // 4 KiB per block-iteration

// Mark modes for the position scan.
#define MODE_NEWLINE 0
#define MODE_TOKEN_START 1

__device__ __forceinline__ bool mark_at(int mode, const u8* text,
                                        long i, long n) {
    u8 c = text[i];
    if (mode == MODE_NEWLINE) return c == '\n';
    // token start: word char whose predecessor is not a word char
    if (!is_word(c)) return false;
    return i == 0 || !is_word(text[i - 1]);
}

// ---------------------------------------------------------------- scan pass 1
// Each block owns a contiguous [base, base + iters*TILE) byte range and
// counts its marks.
// SWAR newline detection: high bit of each byte equal to '\n' (5 VALU
// ops per 4 bytes instead of ~3 per byte) — lines are ~100 bytes, so
// matches are rare and position extraction is off the hot path.
__device__ __forceinline__ u32 nlmask32(u32 w) {
    u32 x = w ^ 0x0A0A0A0Au;
    return (x - 0x01010101u) & ~x & 0x80808080u;
}

__global__ void count_marks_kernel(const u8* __restrict__ text, long n,
                                   int mode, int iters,
                                   u32* __restrict__ counts) {
    long base = (long)blockIdx.x * iters * TILE;
    int tid = threadIdx.x;
    u32 local = 0;
    for (int it = 0; it < iters; ++it) {
        long off = base + (long)it * TILE + (long)tid * VBYTES;
        if (off >= n) break;
        if (off + VBYTES <= n) {
            uint4 v = *reinterpret_cast<const uint4*>(text + off);
            if (mode == MODE_NEWLINE) {
                local += __popc(nlmask32(v.x)) + __popc(nlmask32(v.y))
                       + __popc(nlmask32(v.z)) + __popc(nlmask32(v.w));
            } else {
                const u8* b = reinterpret_cast<const u8*>(&v);
                #pragma unroll
                for (int j = 0; j < VBYTES; ++j) {
                    bool prev_word = (j > 0)
                        ? is_word(b[j - 1])
                        : (off > 0 ? is_word(text[off - 1]) : false);
                    local += is_word(b[j]) && !prev_word;
                }
            }
        } else {
            for (long i = off; i < n; ++i)
                local += mark_at(mode, text, i, n);
        }
    }
    // block reduction: wave reduce then LDS
    __shared__ u32 wsum[BLOCK / WAVE];
    for (int d = WAVE / 2; d > 0; d >>= 1)
        local += __shfl_down(local, d, WAVE);
    int lane = tid & (WAVE - 1), wid = tid / WAVE;
    if (lane == 0) wsum[wid] = local;
    __syncthreads();
    if (tid == 0) {
        u32 total = 0;
        for (int w = 0; w < BLOCK / WAVE; ++w) total += wsum[w];
        counts[blockIdx.x] = total;
    }
}

// ---------------------------------------------------------------- scan pass 2
// Re-scan and write mark positions in ascending order.  Two-phase within
// the block: per-(wave, tile) counts into LDS, ONE barrier, then a
// sync-free scatter pass where every wave derives its own bases from the
// count matrix (the old per-tile cursor + double barrier serialized the
// block on memory latency every 4 KiB).
#define SCAN_MAX_ITERS 16

__device__ __forceinline__ u32 gather_marks(const u8* __restrict__ text,
                                            long n, int mode, long off,
                                            u8* rel) {
    u32 cnt = 0;
    if (off >= n) return 0;
    if (off + VBYTES <= n) {
        uint4 v = *reinterpret_cast<const uint4*>(text + off);
        if (mode == MODE_NEWLINE) {
            const u32 words[4] = {v.x, v.y, v.z, v.w};
            #pragma unroll
            for (int wi = 0; wi < 4; ++wi) {
                u32 m = nlmask32(words[wi]);
                while (m) {
                    int bit = __ffs(m) - 1;
                    rel[cnt++] = (u8)(wi * 4 + (bit >> 3));
                    m &= m - 1;
                }
            }
            return cnt;
        }
        const u8* b = reinterpret_cast<const u8*>(&v);
        #pragma unroll
        for (int j = 0; j < VBYTES; ++j) {
            u8 c = b[j];
            bool prev_word = (j > 0)
                ? is_word(b[j - 1])
                : (off > 0 ? is_word(text[off - 1]) : false);
            bool m = is_word(c) && !prev_word;
            if (m) rel[cnt++] = (u8)j;
        }
    } else {
        long lim = n - off;
        for (long j = 0; j < lim; ++j)
            if (mark_at(mode, text, off + j, n)) rel[cnt++] = (u8)j;
    }
    return cnt;
}

__global__ void write_marks_kernel(const u8* __restrict__ text, long n,
                                   int mode, int iters,
                                   const u32* __restrict__ block_offsets,
                                   u32* __restrict__ out) {
    long base = (long)blockIdx.x * iters * TILE;
    int tid = threadIdx.x;
    int lane = tid & (WAVE - 1), wid = tid / WAVE;
    __shared__ u32 wtile[BLOCK / WAVE][SCAN_MAX_ITERS];

    // phase 1: per-(wave, tile) mark counts
    for (int it = 0; it < iters; ++it) {
        long off = base + (long)it * TILE + (long)tid * VBYTES;
        u8 rel[VBYTES];
        u32 cnt = gather_marks(text, n, mode, off, rel);
        u32 tot = cnt;
        for (int d = WAVE / 2; d > 0; d >>= 1)
            tot += __shfl_down(tot, d, WAVE);
        if (lane == 0) wtile[wid][it] = tot;
    }
    __syncthreads();

    // phase 2: rescan (L2-hot) and scatter; bases derived per wave
    u32 run = block_offsets[blockIdx.x];
    for (int it = 0; it < iters; ++it) {
        u32 wb = run;
        u32 tile_tot = 0;
        for (int w = 0; w < BLOCK / WAVE; ++w) {
            if (w < wid) wb += wtile[w][it];
            tile_tot += wtile[w][it];
        }
        long off = base + (long)it * TILE + (long)tid * VBYTES;
        u8 rel[VBYTES];
        u32 cnt = gather_marks(text, n, mode, off, rel);
        u32 scan = cnt;
        for (int d = 1; d < WAVE; d <<= 1) {
            u32 x = __shfl_up(scan, d, WAVE);
            if (lane >= d) scan += x;
        }
        u32 excl = wb + scan - cnt;
        for (u32 j = 0; j < cnt; ++j)
            out[excl + j] = (u32)(off + rel[j]);
        run += tile_tot;
    }
}

// ------------------------------------------------------------ tokenize+count
// One thread per token: hash the token (K1), locate its document (binary
// search over newline positions), dedupe (doc, token) in the seen table
// and bump the document-frequency table (K6).  First global occurrence of
// a token also records (byte offset, length) in the string dictionary so
// results can be materialized as text.
__global__ void tfidf_count_kernel(
        const u8* __restrict__ text, long n,
        const u32* __restrict__ nl_pos, long n_nl,
        const u32* __restrict__ tok_start, long n_tok,
        u64* __restrict__ seen_keys, u64 seen_mask,
        u64* __restrict__ cnt_keys, u64* __restrict__ cnt_vals,
        u64 cnt_mask,
        u64* __restrict__ dict_keys, u64* __restrict__ dict_vals,
        u64 dict_mask, u64 pos_base, u64 doc_base) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n_tok;
         t += stride) {
        u32 start = tok_start[t];
        long p = start;
        u64 h = 0;
        u32 j = 0;
        while (p < n) {
            u8 c = text[p];
            if (!is_word(c)) break;
            h = tokmix_step(h, j, lower_ascii(c));
            ++p;
            ++j;
        }
        u32 len = (u32)(p - start);
        h = tokmix_final(h, len);
        // doc id = number of newlines strictly before `start`
        long lo = 0, hi = n_nl;
        while (lo < hi) {
            long mid = (lo + hi) >> 1;
            if (nl_pos[mid] < start) lo = mid + 1; else hi = mid;
        }
        u64 doc = (u64)lo + doc_base;
        u64 tok_key = h ? h : 1ULL;
        u64 sk = splitmix64(h ^ (doc * 0x9E3779B97F4A7C15ULL));
        if (!sk) sk = 1;
        u64 slot;
        if (table_insert_u64(seen_keys, seen_mask, sk, &slot)) {
            table_add_u64(cnt_keys, cnt_vals, cnt_mask, tok_key, 1ULL);
            if (table_insert_u64(dict_keys, dict_mask, tok_key, &slot))
                dict_vals[slot] = ((pos_base + (u64)start) << 8)
                                  | (u64)min(len, 255u);
        }
    }
}

// ------------------------------------------------------- doc-centric count
// Fast path: one wave per document.  The line is staged into LDS with
// coalesced loads, token starts are found in-register, and the per-doc
// `set()` dedupe is a tiny per-wave LDS hash set — no global seen table,
// no per-token binary search, one pass over the text.  Documents whose
// distinct-token count overflows the LDS set fall back to a global
// (doc,hash)-keyed seen table; if THAT overflows its probe bound the
// kernel sets an error flag and the host reruns the chunk on the fully
// general token-centric kernel above.
#ifndef DOC_WAVES
#define DOC_WAVES 4                  // waves per block
#endif
#ifndef STAGE_B
#define STAGE_B 2048                 // staged line segment bytes
#endif
#define SEG_OVERLAP 272              // > max dict token length (255)
#ifndef DOC_SET
#define DOC_SET 256                  // per-wave dedupe set slots (pow2)
#endif
#ifndef FB_PROBE_CAP
#define FB_PROBE_CAP 512
#endif
#ifndef DOC_BLK
#define DOC_BLK 64                   // contiguous docs per wave block
#endif
#ifndef GROUP_BYTES
#define GROUP_BYTES 1024             // target bytes per doc group
#endif
#ifndef CCACHE
#define CCACHE 1024                  // block-level LDS count cache slots
#endif

// Two-level counting: Zipf-hot keys would serialize ~50M same-address L2
// atomics; the block-level LDS cache turns that into one global add per
// (block, hot key).  Cache misses (cold keys) go straight to the global
// table — cold keys have no contention.
__device__ __forceinline__ void block_count_add(
        u64* __restrict__ cck, u32* __restrict__ ccv, u64 key,
        u64* __restrict__ gk, u64* __restrict__ gv, u64 gmask) {
    u32 slot = (u32)(key & (CCACHE - 1));
    for (int probe = 0; probe < 4; ++probe) {
        u64 cur = cck[slot];
        if (cur == key) { atomicAdd(&ccv[slot], 1u); return; }
        if (cur == 0ULL) {
            u64 prev = atomicCAS(&cck[slot], 0ULL, key);
            if (prev == 0ULL || prev == key) {
                atomicAdd(&ccv[slot], 1u);
                return;
            }
        }
        slot = (slot + 1) & (CCACHE - 1);
    }
    table_add_u64(gk, gv, gmask, key, 1ULL);
}

__device__ __forceinline__ int lds_set_insert(u64* set, u64 h) {
    // 1 = fresh, 0 = dup, -1 = set full (h definitely absent: full scan)
    u32 slot = (u32)(h & (DOC_SET - 1));
    for (int probe = 0; probe < DOC_SET; ++probe) {
        u64 prev = atomicCAS(&set[slot], 0ULL, h);
        if (prev == 0ULL) return 1;
        if (prev == h) return 0;
        slot = (slot + 1) & (DOC_SET - 1);
    }
    return -1;
}

__global__ void __launch_bounds__(DOC_WAVES * WAVE, 5)
tfidf_docs_kernel(const u8* __restrict__ text, long n,
                  const u32* __restrict__ nl_pos, long n_nl, long n_docs,
                  u64* __restrict__ cnt_keys, u64* __restrict__ cnt_vals,
                  u64 cnt_mask,
                  u64* __restrict__ dict_keys, u64* __restrict__ dict_vals,
                  u64 dict_mask, u64 pos_base,
                  u64* __restrict__ fb_seen, u64 fb_mask,
                  u32* __restrict__ err_flag, u32 ablate) {
    // v3: group-of-docs stream processing, all 64 lanes on every byte.
    //  * a wave owns DOC_BLK contiguous docs; one coalesced nl_pos vector
    //    load gives all their boundaries (shfl per query)
    //  * docs are processed in *groups* (~GROUP_BYTES of consecutive
    //    docs): windows run continuously across doc boundaries, so short
    //    docs no longer waste partial windows or per-doc framing
    //  * doc identity inside a window comes from the newline ballot;
    //    dedupe keys are salted with the doc id, so the per-wave LDS set
    //    only needs clearing once per group (always at a doc boundary)
    //  * token hashes by wave-wide segmented XOR scan (tokmix); tokens
    //    crossing window/stage edges ride a wave-uniform carry
    __shared__ __align__(16) u8 stage[DOC_WAVES][STAGE_B + 16];
    __shared__ u64 dset[DOC_WAVES][DOC_SET];
    __shared__ u64 cck[CCACHE];
    __shared__ u32 ccv[CCACHE];
    __shared__ u64 toktab[256];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const long gwave = (long)blockIdx.x * DOC_WAVES + wid;
    const long nwaves = (long)gridDim.x * DOC_WAVES;
    u8* st = stage[wid];
    u64* set = dset[wid];

    for (int i = threadIdx.x; i < CCACHE; i += blockDim.x) {
        cck[i] = 0;
        ccv[i] = 0;
    }
    for (int i = threadIdx.x; i < 256; i += blockDim.x)
        toktab[i] = toktab_entry((u8)i);
    __syncthreads();

    for (long dbase = gwave * DOC_BLK; dbase < n_docs;
         dbase += nwaves * DOC_BLK) {
        const long dlim = min(dbase + (long)DOC_BLK, n_docs);
        const long di = dbase + lane;
        // lane i holds the END byte (newline pos, or n) of doc dbase+i
        const u32 nl_lane = (di < n_nl) ? nl_pos[di] : (u32)n;
        long win_lo = -1, win_hi = -1, aseg = 0;
        long gd = dbase;                       // first doc of the group
        long gs = dbase ? (long)nl_pos[dbase - 1] + 1 : 0;

        while (gd < dlim) {
            // group = docs [gd, ge): consecutive docs spanning at most
            // GROUP_BYTES (a longer single doc forms its own group).
            // ends[lane] = end byte of doc dbase+lane (exclusive \n).
            const u64 fit = __ballot(
                di < dlim && di >= gd
                && (long)nl_lane <= gs + (long)GROUP_BYTES);
            long ge;
            if (fit) {
                ge = dbase + 63 - __clzll(fit) + 1;
            } else {
                ge = gd + 1;                   // oversized doc: alone
            }
            const long gend = (ge - 1 < n_nl)
                ? (long)(u32)__shfl((int)nl_lane, (int)(ge - 1 - dbase),
                                    WAVE)
                : n;

            for (int s = lane; s < DOC_SET; s += WAVE) set[s] = 0;
            u32 carry_word = 0;                // wave-uniform token carry
            u32 carry_len = 0;
            u64 carry_g = 0;
            long carry_start = gs;
            u32 carry_lines = 0;               // newlines seen in group

            // dedupe + count + dict insert for one finished token
            auto emit_token = [&](u64 gh, u32 tl, long tstart,
                                  long doc_id) {
                if (ablate & 1) {              // ablation: consume hash
                    if (gh == 0xdeadbeefdeadbeefULL) err_flag[1] = 1;
                    return;
                }
                const u64 hh = tokmix_final(gh, tl);
                const u64 key = hh ? hh : 1ULL;
                u64 dk = hh ^ splitmix64((u64)doc_id + 0x5bd1e995ULL);
                if (!dk) dk = 1;
                int fresh = lds_set_insert(set, dk);
                if (fresh < 0) {
                    // set overflow: global (doc,hash) seen fallback
                    u64 sk = splitmix64(hh ^ ((u64)doc_id
                                              * 0x9E3779B97F4A7C15ULL));
                    if (!sk) sk = 1;
                    u64 slot = sk & fb_mask;
                    fresh = 0;
                    int probe = 0;
                    while (true) {
                        u64 prev = atomicCAS(&fb_seen[slot], 0ULL, sk);
                        if (prev == 0ULL) { fresh = 1; break; }
                        if (prev == sk) break;
                        slot = (slot + 1) & fb_mask;
                        if (++probe > FB_PROBE_CAP) {
                            atomicOr(err_flag, 1u);
                            break;
                        }
                    }
                }
                if (fresh == 1 && !(ablate & 2)) {
                    block_count_add(cck, ccv, key, cnt_keys, cnt_vals,
                                    cnt_mask);
                    u64 slot;
                    if (!(ablate & 4)
                        && table_insert_u64(dict_keys, dict_mask, key,
                                            &slot))
                        dict_vals[slot] =
                            ((pos_base + (u64)tstart) << 8)
                            | (u64)min(tl, 255u);
                }
            };

            for (long seg = gs; seg < gend; ) {
                if (seg < win_lo || seg >= win_hi) {
                    // (re)stage an aligned window from seg; bytes beyond
                    // this group belong to following docs and are
                    // reused.  Full interior windows use the async
                    // global->LDS copy (zero staging VGPRs, no
                    // reg->ds_write pass; LDS dest = wave-uniform base
                    // + lane*16 = exactly this linear layout).  A
                    // double-buffered prefetch variant measured WORSE
                    // (999M vs 1130M rows/s): the extra 8 KB LDS costs
                    // an occupancy step, which outweighs hiding the
                    // staging stall.  Corpus tail keeps the byte path.
                    aseg = seg & ~15L;
                    const int stage_bytes =
                        (int)min((long)(STAGE_B + 16), n - aseg);
                    if (aseg + STAGE_B + 16 <= n) {
                        #pragma unroll
                        for (int i = 0; i < STAGE_B; i += WAVE * 16)
                            __builtin_amdgcn_global_load_lds(
                                (const u32*)(text + aseg + i
                                             + lane * 16),
                                (u32*)(st + i), 16, 0, 0);
                        if (lane == 0)
                            __builtin_amdgcn_global_load_lds(
                                (const u32*)(text + aseg + STAGE_B),
                                (u32*)(st + STAGE_B), 16, 0, 0);
                    } else {
                        for (int i = lane * 16; i < stage_bytes;
                             i += WAVE * 16) {
                            if (aseg + i + 16 <= n) {
                                *reinterpret_cast<uint4*>(st + i) =
                                    *reinterpret_cast<const uint4*>(
                                        text + aseg + i);
                            } else {
                                for (int j = i; j < stage_bytes; ++j)
                                    st[j] = (aseg + j < n)
                                        ? text[aseg + j] : (u8)0;
                            }
                        }
                    }
                    asm volatile(
                        "s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
                    __builtin_amdgcn_wave_barrier();
                    win_lo = seg;
                    win_hi = aseg + stage_bytes;
                }
                const long seg_end = min(gend, win_hi);
                const int seg_len = (int)(seg_end - seg);
                const int soff = (int)(seg - aseg);
                const bool group_continues = seg_end < gend;
                const int nwin = (seg_len + WAVE - 1) / WAVE;
                for (int wdx = 0; wdx < nwin; ++wdx) {
                    const int wb = wdx * WAVE;
                    const int p = wb + lane;
                    const bool valid = p < seg_len;
                    const u8 c = valid ? st[soff + p] : (u8)0;
                    const bool w = valid && is_word(c);
                    const u64 wm = __ballot(w);
                    const u64 nlm = __ballot(valid && c == '\n');
                    if (carry_word && !(wm & 1ULL)) {
                        // carried token ended exactly at the window edge
                        if (lane == 0)
                            emit_token(carry_g, carry_len, carry_start,
                                       gd + carry_lines);
                        carry_word = 0;
                        carry_len = 0;
                        carry_g = 0;
                    }
                    const u64 sm = wm & ~((wm << 1) | (u64)carry_word);
                    const u64 below_inc = (lane == 63)
                        ? ~0ULL : ((1ULL << (lane + 1)) - 1ULL);
                    const u64 sm_le = sm & below_inc;
                    int s = WAVE;          // token start (window coords)
                    int pos = 0;           // byte position within token
                    u64 g = 0;
                    if (w) {
                        s = sm_le ? (63 - __clzll(sm_le))
                                  : -(int)carry_len;
                        pos = lane - s;
                        g = rotl64(toktab[lower_ascii(c)], (u32)pos);
                    }
                    // segmented inclusive XOR scan; <8-byte tokens (the
                    // common case) need only the first three steps
                    #pragma unroll
                    for (int dsh = 1; dsh <= 4; dsh <<= 1) {
                        const u64 g2 = __shfl_up(g, dsh, WAVE);
                        if (w && lane >= dsh && (lane - dsh) >= s)
                            g ^= g2;
                    }
                    if (__ballot(w && pos >= 8)) {
                        #pragma unroll
                        for (int dsh = 8; dsh < WAVE; dsh <<= 1) {
                            const u64 g2 = __shfl_up(g, dsh, WAVE);
                            if (w && lane >= dsh && (lane - dsh) >= s)
                                g ^= g2;
                        }
                    }
                    if (w && s < 0) g ^= carry_g;  // continuing prefix

                    const int last_valid = min(seg_len - wb, WAVE) - 1;
                    bool at_end = w
                        && (lane == 63 ? true
                                       : !((wm >> (lane + 1)) & 1));
                    if (lane == last_valid && w
                        && (last_valid == 63 || group_continues))
                        at_end = false;    // may continue: carry it

                    if (at_end) {
                        const u64 below = (1ULL << lane) - 1ULL;
                        const long doc_id = gd + carry_lines
                            + __popcll(nlm & below);
                        emit_token(g, (u32)(pos + 1),
                                   (s >= 0) ? (seg + wb + s)
                                            : carry_start, doc_id);
                    }

                    // wave-uniform carry update from the tail lane
                    const int t_w = (int)((wm >> last_valid) & 1);
                    const int t_end = __shfl((int)at_end, last_valid,
                                             WAVE);
                    if (t_w && !t_end) {
                        const int t_pos = __shfl(pos, last_valid, WAVE);
                        const int t_s = __shfl(s, last_valid, WAVE);
                        const u64 t_g = __shfl(g, last_valid, WAVE);
                        carry_word = 1;
                        carry_len = (u32)(t_pos + 1);
                        carry_g = t_g;
                        if (t_s >= 0) carry_start = seg + wb + t_s;
                    } else {
                        carry_word = 0;
                        carry_len = 0;
                        carry_g = 0;
                    }
                    carry_lines += (u32)__popcll(nlm);
                }
                __builtin_amdgcn_wave_barrier();
                seg = seg_end;
            }
            if (carry_word) {
                // group ended at a window edge with a live token
                if (lane == 0)
                    emit_token(carry_g, carry_len, carry_start,
                               gd + carry_lines);
            }
            gd = ge;
            gs = gend + 1;                     // past the newline
        }
    }

    // flush the block's count cache
    __syncthreads();
    for (int i = threadIdx.x; i < CCACHE; i += blockDim.x)
        if (cck[i])
            table_add_u64(cnt_keys, cnt_vals, cnt_mask, cck[i],
                          (u64)ccv[i]);
}

// ---------------------------------------------------------------- table ops
__global__ void table_merge_kernel(const u64* __restrict__ in_keys,
                                   const long* __restrict__ in_vals,
                                   long n, u64* __restrict__ keys,
                                   u64* __restrict__ vals, u64 mask) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u64 k = in_keys[i];
        if (k) table_add_u64(keys, vals, mask, k, (u64)in_vals[i]);
    }
}

__global__ void table_put_kernel(const u64* __restrict__ in_keys,
                                 const u64* __restrict__ in_vals, long n,
                                 u64* __restrict__ keys,
                                 u64* __restrict__ vals, u64 mask) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u64 k = in_keys[i];
        if (!k) continue;
        u64 slot;
        if (table_insert_u64(keys, mask, k, &slot))
            vals[slot] = in_vals[i];
    }
}

__global__ void table_extract_kernel(const u64* __restrict__ keys,
                                     const u64* __restrict__ vals,
                                     long cap, u64* __restrict__ out_k,
                                     long* __restrict__ out_v,
                                     u64* __restrict__ cursor) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < cap;
         i += stride) {
        u64 k = keys[i];
        if (k) {
            u64 j = atomicAdd(cursor, 1ULL);
            out_k[j] = k;
            out_v[j] = (long)vals[i];
        }
    }
}

__global__ void table_lookup_kernel(const u64* __restrict__ keys,
                                    const u64* __restrict__ vals, u64 mask,
                                    const u64* __restrict__ query, long n,
                                    u64* __restrict__ out) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u64 k = query[i];
        u64 slot = k & mask;
        u64 r = 0;
        while (true) {
            u64 cur = keys[slot];
            if (cur == k) { r = vals[slot]; break; }
            if (cur == 0) break;
            slot = (slot + 1) & mask;
        }
        out[i] = r;
    }
}

// ------------------------------------------------------------------ epilogue
// idf = log(1 + total/df): the reference's cross_right scalar apply (K9 +
// benchmarks/tf-idf-dampr.py:18-20), fused elementwise.
__global__ void idf_kernel(const long* __restrict__ df, long n,
                           double total, double* __restrict__ out) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        out[i] = log(1.0 + total / (double)df[i]);
}

// Gather token strings: out[offsets[i] .. offsets[i]+len_i) = bytes of
// token i (packed = pos<<8 | len).
__global__ void gather_tokens_kernel(const u8* __restrict__ text,
                                     const u64* __restrict__ packed, long n,
                                     const long* __restrict__ offsets,
                                     u8* __restrict__ out) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u64 pk = packed[i];
        u32 pos = (u32)(pk >> 8);
        u32 len = (u32)(pk & 0xFF);
        long o = offsets[i];
        for (u32 j = 0; j < len; ++j)
            out[o + j] = lower_ascii(text[pos + j]);
    }
}

// ------------------------------------------------------------------ tsv sink
// Device-side TSV formatting (K10's "serialize results" role): rows are
// "token\tDF\tIDF\n" with IDF fixed at 9 decimals.  Two passes: sizes
// (host cumsums them) then formatted writes.  Replaces a per-row Python
// format loop that dominated step time.

__device__ __forceinline__ int dec_digits_u64(u64 v) {
    int d = 1;
    while (v >= 10) { v /= 10; ++d; }
    return d;
}

__device__ __forceinline__ void fmt_idf_parts(double x, u64* int_part,
                                              u64* frac_part) {
    double scaled = x * 1e9 + 0.5;
    u64 v = (u64)scaled;
    *int_part = v / 1000000000ULL;
    *frac_part = v % 1000000000ULL;
}

__global__ void tsv_sizes_kernel(const long* __restrict__ lens,
                                 const long* __restrict__ df,
                                 const double* __restrict__ idf, long n,
                                 long* __restrict__ sizes) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u64 ip, fp;
        fmt_idf_parts(idf[i], &ip, &fp);
        sizes[i] = lens[i] + 1 + dec_digits_u64((u64)df[i]) + 1
                 + dec_digits_u64(ip) + 1 + 9 + 1;
    }
}

__device__ __forceinline__ u8* write_u64_dec(u8* p, u64 v, int width) {
    for (int j = width - 1; j >= 0; --j) { p[j] = '0' + (v % 10); v /= 10; }
    return p + width;
}

__global__ void tsv_format_kernel(const u8* __restrict__ blob,
                                  const long* __restrict__ tok_off,
                                  const long* __restrict__ lens,
                                  const long* __restrict__ df,
                                  const double* __restrict__ idf,
                                  const long* __restrict__ row_off, long n,
                                  u8* __restrict__ out) {
    long stride = (long)gridDim.x * blockDim.x;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        u8* p = out + row_off[i];
        const u8* t = blob + tok_off[i];
        for (long j = 0; j < lens[i]; ++j) *p++ = t[j];
        *p++ = '\t';
        u64 d = (u64)df[i];
        p = write_u64_dec(p, d, dec_digits_u64(d));
        *p++ = '\t';
        u64 ip, fp;
        fmt_idf_parts(idf[i], &ip, &fp);
        p = write_u64_dec(p, ip, dec_digits_u64(ip));
        *p++ = '.';
        p = write_u64_dec(p, fp, 9);
        *p++ = '\n';
    }
}

// ==========================================================================
// Host wrappers
// ==========================================================================

namespace {

inline hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

inline int grid_for(long work, int block = BLOCK, int cap = 4096) {
    long g = (work + block - 1) / block;
    return (int)std::min<long>(std::max<long>(g, 1), cap);
}

void check_u8(const torch::Tensor& t) {
    TORCH_CHECK(t.is_cuda() && t.dtype() == torch::kUInt8 &&
                t.is_contiguous(), "expected contiguous u8 device tensor");
}

constexpr int SCAN_ITERS = 8;   // 32 KiB per block

}  // namespace

// Returns per-block mark counts; python computes exclusive offsets.
torch::Tensor mark_counts(torch::Tensor text, long mode) {
    check_u8(text);
    long n = text.numel();
    long span = (long)SCAN_ITERS * TILE;
    long nblocks = (n + span - 1) / span;
    auto counts = torch::empty({std::max(nblocks, 1L)},
        torch::TensorOptions().dtype(torch::kUInt32).device(text.device()));
    if (n == 0) { counts.zero_(); return counts; }
    hipLaunchKernelGGL(count_marks_kernel, dim3((u32)nblocks), dim3(BLOCK),
                       0, cur_stream(), text.data_ptr<u8>(), n, (int)mode,
                       SCAN_ITERS, (u32*)counts.data_ptr());
    return counts;
}

void mark_positions(torch::Tensor text, long mode,
                    torch::Tensor block_offsets, torch::Tensor out) {
    check_u8(text);
    long n = text.numel();
    if (n == 0 || out.numel() == 0) return;
    long span = (long)SCAN_ITERS * TILE;
    long nblocks = (n + span - 1) / span;
    hipLaunchKernelGGL(write_marks_kernel, dim3((u32)nblocks), dim3(BLOCK),
                       0, cur_stream(), text.data_ptr<u8>(), n, (int)mode,
                       SCAN_ITERS, (u32*)block_offsets.data_ptr(),
                       (u32*)out.data_ptr());
}

void tfidf_count(torch::Tensor text, torch::Tensor nl_pos,
                 torch::Tensor tok_start, torch::Tensor seen_keys,
                 torch::Tensor cnt_keys, torch::Tensor cnt_vals,
                 torch::Tensor dict_keys, torch::Tensor dict_vals,
                 long pos_base, long doc_base) {
    check_u8(text);
    long n_tok = tok_start.numel();
    if (n_tok == 0) return;
    hipLaunchKernelGGL(tfidf_count_kernel,
        dim3(grid_for(n_tok)), dim3(BLOCK), 0, cur_stream(),
        text.data_ptr<u8>(), text.numel(),
        (const u32*)nl_pos.data_ptr(), nl_pos.numel(),
        (const u32*)tok_start.data_ptr(), n_tok,
        (u64*)seen_keys.data_ptr(), (u64)(seen_keys.numel() - 1),
        (u64*)cnt_keys.data_ptr(), (u64*)cnt_vals.data_ptr(),
        (u64)(cnt_keys.numel() - 1),
        (u64*)dict_keys.data_ptr(), (u64*)dict_vals.data_ptr(),
        (u64)(dict_keys.numel() - 1), (u64)pos_base, (u64)doc_base);
}

// Returns 0 on success, 1 when the fallback seen table overflowed (host
// must rerun the chunk via the token-centric kernel).
long tfidf_count_docs(torch::Tensor text, torch::Tensor nl_pos,
                      long n_docs, torch::Tensor cnt_keys,
                      torch::Tensor cnt_vals, torch::Tensor dict_keys,
                      torch::Tensor dict_vals, long pos_base,
                      torch::Tensor fb_seen, torch::Tensor err_flag,
                      long ablate) {
    check_u8(text);
    if (n_docs == 0) return 0;
    long waves_needed = (n_docs + 63) / 64;     // DOC_BLK docs per wave
    long blocks = std::min<long>((waves_needed + DOC_WAVES - 1) / DOC_WAVES,
                                 8192);
    hipLaunchKernelGGL(tfidf_docs_kernel, dim3((u32)blocks),
        dim3(DOC_WAVES * WAVE), 0, cur_stream(),
        text.data_ptr<u8>(), text.numel(),
        (const u32*)nl_pos.data_ptr(), nl_pos.numel(), n_docs,
        (u64*)cnt_keys.data_ptr(), (u64*)cnt_vals.data_ptr(),
        (u64)(cnt_keys.numel() - 1),
        (u64*)dict_keys.data_ptr(), (u64*)dict_vals.data_ptr(),
        (u64)(dict_keys.numel() - 1), (u64)pos_base,
        (u64*)fb_seen.data_ptr(), (u64)(fb_seen.numel() - 1),
        (u32*)err_flag.data_ptr(), (u32)ablate);
    return 0;
}

void table_merge(torch::Tensor in_keys, torch::Tensor in_vals,
                 torch::Tensor keys, torch::Tensor vals) {
    long n = in_keys.numel();
    if (n == 0) return;
    hipLaunchKernelGGL(table_merge_kernel, dim3(grid_for(n)), dim3(BLOCK),
        0, cur_stream(), (const u64*)in_keys.data_ptr(),
        in_vals.data_ptr<long>(), n, (u64*)keys.data_ptr(),
        (u64*)vals.data_ptr(), (u64)(keys.numel() - 1));
}

void table_put(torch::Tensor in_keys, torch::Tensor in_vals,
               torch::Tensor keys, torch::Tensor vals) {
    long n = in_keys.numel();
    if (n == 0) return;
    hipLaunchKernelGGL(table_put_kernel, dim3(grid_for(n)), dim3(BLOCK),
        0, cur_stream(), (const u64*)in_keys.data_ptr(),
        (const u64*)in_vals.data_ptr(), n, (u64*)keys.data_ptr(),
        (u64*)vals.data_ptr(), (u64)(keys.numel() - 1));
}

std::vector<torch::Tensor> table_extract(torch::Tensor keys,
                                         torch::Tensor vals, long n_out) {
    auto dev = keys.device();
    auto out_k = torch::empty({n_out},
        torch::TensorOptions().dtype(torch::kInt64).device(dev));
    auto out_v = torch::empty({n_out},
        torch::TensorOptions().dtype(torch::kInt64).device(dev));
    auto cursor = torch::zeros({1},
        torch::TensorOptions().dtype(torch::kInt64).device(dev));
    long cap = keys.numel();
    if (cap > 0 && n_out > 0)
        hipLaunchKernelGGL(table_extract_kernel, dim3(grid_for(cap)),
            dim3(BLOCK), 0, cur_stream(), (const u64*)keys.data_ptr(),
            (const u64*)vals.data_ptr(), cap, (u64*)out_k.data_ptr(),
            out_v.data_ptr<long>(), (u64*)cursor.data_ptr());
    return {out_k, out_v, cursor};
}

torch::Tensor table_lookup(torch::Tensor keys, torch::Tensor vals,
                           torch::Tensor query) {
    long n = query.numel();
    auto out = torch::zeros({n},
        torch::TensorOptions().dtype(torch::kInt64).device(query.device()));
    if (n > 0)
        hipLaunchKernelGGL(table_lookup_kernel, dim3(grid_for(n)),
            dim3(BLOCK), 0, cur_stream(), (const u64*)keys.data_ptr(),
            (const u64*)vals.data_ptr(), (u64)(keys.numel() - 1),
            (const u64*)query.data_ptr(), n, (u64*)out.data_ptr());
    return out;
}

torch::Tensor idf(torch::Tensor df, double total) {
    long n = df.numel();
    auto out = torch::empty({n},
        torch::TensorOptions().dtype(torch::kFloat64).device(df.device()));
    if (n > 0)
        hipLaunchKernelGGL(idf_kernel, dim3(grid_for(n)), dim3(BLOCK), 0,
            cur_stream(), df.data_ptr<long>(), n, total,
            out.data_ptr<double>());
    return out;
}

torch::Tensor gather_tokens(torch::Tensor text, torch::Tensor packed,
                            torch::Tensor offsets, long total_bytes) {
    check_u8(text);
    long n = packed.numel();
    auto out = torch::empty({std::max(total_bytes, 1L)},
        torch::TensorOptions().dtype(torch::kUInt8).device(text.device()));
    if (n > 0)
        hipLaunchKernelGGL(gather_tokens_kernel, dim3(grid_for(n)),
            dim3(BLOCK), 0, cur_stream(), text.data_ptr<u8>(),
            (const u64*)packed.data_ptr(), n, offsets.data_ptr<long>(),
            out.data_ptr<u8>());
    return out;
}

torch::Tensor tsv_sizes(torch::Tensor lens, torch::Tensor df,
                        torch::Tensor idf) {
    long n = lens.numel();
    auto out = torch::empty({std::max(n, 1L)},
        torch::TensorOptions().dtype(torch::kInt64).device(lens.device()));
    if (n > 0)
        hipLaunchKernelGGL(tsv_sizes_kernel, dim3(grid_for(n)), dim3(BLOCK),
            0, cur_stream(), lens.data_ptr<long>(), df.data_ptr<long>(),
            idf.data_ptr<double>(), n, out.data_ptr<long>());
    return out;
}

torch::Tensor tsv_format(torch::Tensor blob, torch::Tensor tok_off,
                         torch::Tensor lens, torch::Tensor df,
                         torch::Tensor idf, torch::Tensor row_off,
                         long total_bytes) {
    long n = lens.numel();
    auto out = torch::empty({std::max(total_bytes, 1L)},
        torch::TensorOptions().dtype(torch::kUInt8).device(blob.device()));
    if (n > 0)
        hipLaunchKernelGGL(tsv_format_kernel, dim3(grid_for(n)),
            dim3(BLOCK), 0, cur_stream(), blob.data_ptr<u8>(),
            tok_off.data_ptr<long>(), lens.data_ptr<long>(),
            df.data_ptr<long>(), idf.data_ptr<double>(),
            row_off.data_ptr<long>(), n, out.data_ptr<u8>());
    return out;
}

// Implemented in dampr_sort.hip
torch::Tensor rs_hist(torch::Tensor keys, long shift, long nblocks);
torch::Tensor rs_digit_fold(torch::Tensor keys);
void rs_scatter(torch::Tensor keys, torch::Tensor payload,
                torch::Tensor scanned, long shift, long nblocks,
                torch::Tensor out_k, torch::Tensor out_p);
void seg_reduce(torch::Tensor seg, torch::Tensor vals, torch::Tensor out,
                long op);
void hj_build(torch::Tensor keys_r, torch::Tensor t_keys,
              torch::Tensor t_head, torch::Tensor next);
torch::Tensor hj_count(torch::Tensor keys_l, torch::Tensor t_keys,
                       torch::Tensor t_head, torch::Tensor next,
                       long left_outer);
long rs_span();
torch::Tensor seg_count(torch::Tensor keys);
void seg_reduce_fused(torch::Tensor keys, torch::Tensor vals,
                      torch::Tensor tile_base, long op,
                      torch::Tensor out_keys, torch::Tensor out_vals);
void mp_merge(torch::Tensor ka, torch::Tensor pa, torch::Tensor kb,
              torch::Tensor pb, long bias_signed, torch::Tensor out_k,
              torch::Tensor out_p);
void varlen_gather(torch::Tensor blob, torch::Tensor src_off,
                   torch::Tensor lens, torch::Tensor new_offs,
                   torch::Tensor out);
std::vector<torch::Tensor> hj_emit(torch::Tensor keys_l,
                                   torch::Tensor t_keys,
                                   torch::Tensor t_head,
                                   torch::Tensor next,
                                   torch::Tensor offsets, long total,
                                   long left_outer, long track_matched,
                                   long nr);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rs_hist", &rs_hist, "radix pass histogram (bin-major)");
    m.def("rs_digit_fold", &rs_digit_fold,
          "AND/OR fold over keys (constant-digit skip detection)");
    m.def("rs_scatter", &rs_scatter, "stable radix scatter pass");
    m.def("seg_reduce", &seg_reduce,
          "segmented reduce over sorted runs (op 0=sum,1=min,2=max)");
    m.def("hj_build", &hj_build, "hash-join build (chained)");
    m.def("varlen_gather", &varlen_gather,
          "row gather for byte-arena value columns");
    m.def("rs_span", &rs_span,
          "elements per radix block (dampr_sort.hip RS_SPAN)");
    m.def("seg_count", &seg_count,
          "segment-start counts per tile (fused group-by, pass 1)");
    m.def("seg_reduce_fused", &seg_reduce_fused,
          "fused boundary-derive + segmented reduce over sorted keys");
    m.def("mp_merge", &mp_merge,
          "stable merge-path 2-way merge of sorted (key, payload) runs");
    m.def("hj_count", &hj_count, "hash-join probe match counts");
    m.def("hj_emit", &hj_emit, "hash-join emit (l,r) row-index pairs");
    m.def("tsv_sizes", &tsv_sizes, "per-row TSV byte sizes");
    m.def("tsv_format", &tsv_format, "format token/df/idf rows as TSV");
    m.def("mark_counts", &mark_counts,
          "per-block counts of marks (mode 0=newline, 1=token start)");
    m.def("mark_positions", &mark_positions,
          "write ascending mark positions given exclusive block offsets");
    m.def("tfidf_count", &tfidf_count,
          "tokenize+hash+per-doc-dedupe+df-count in one pass");
    m.def("tfidf_count_docs", &tfidf_count_docs,
          "wave-per-doc tokenize+dedupe+count (fast path)");
    m.def("table_merge", &table_merge, "add (k,v) pairs into a hash table");
    m.def("table_put", &table_put,
          "insert (k,v) pairs if absent (first writer wins)");
    m.def("table_extract", &table_extract,
          "compact non-empty table slots to (keys, vals, count)");
    m.def("table_lookup", &table_lookup, "probe table for query keys");
    m.def("idf", &idf, "idf = log(1 + total/df)");
    m.def("gather_tokens", &gather_tokens,
          "materialize token byte strings from dict entries");
}
