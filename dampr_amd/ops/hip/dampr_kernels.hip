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

// ------------------------------------------------- single-pass positions
// Decoupled-lookback stream compaction (rocprim-style): one pass over the
// text produces the ordered mark-position array and the total, replacing
// the old count+write two-pass (two full text reads + host cumsum).
// status[b] packs {flag:2, value:62}: 1 = aggregate ready, 2 = inclusive
// prefix ready.

#define MK_FLAG_AGG  (1ULL << 62)
#define MK_FLAG_INC  (2ULL << 62)
#define MK_VAL(x)    ((x) & ((1ULL << 62) - 1))

__device__ __forceinline__ u32 gather_marks(const u8* __restrict__ text,
                                            long n, int mode, long off,
                                            u8* rel) {
    u32 cnt = 0;
    if (off >= n) return 0;
    if (off + VBYTES <= n) {
        uint4 v = *reinterpret_cast<const uint4*>(text + off);
        const u8* b = reinterpret_cast<const u8*>(&v);
        #pragma unroll
        for (int j = 0; j < VBYTES; ++j) {
            u8 c = b[j];
            bool m;
            if (mode == MODE_NEWLINE) {
                m = (c == '\n');
            } else {
                bool prev_word = (j > 0)
                    ? is_word(b[j - 1])
                    : (off > 0 ? is_word(text[off - 1]) : false);
                m = is_word(c) && !prev_word;
            }
            if (m) rel[cnt++] = (u8)j;
        }
    } else {
        long lim = n - off;
        for (long j = 0; j < lim; ++j)
            if (mark_at(mode, text, off + j, n)) rel[cnt++] = (u8)j;
    }
    return cnt;
}

#define MK_ITERS 16                   // tiles per block (64 KiB)

__global__ void marks_lookback_kernel(const u8* __restrict__ text, long n,
                                      int mode,
                                      volatile u64* __restrict__ status,
                                      u32* __restrict__ out,
                                      u64* __restrict__ total) {
    const long b = blockIdx.x;
    const long off0 = b * (long)MK_ITERS * TILE;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1), wid = tid / WAVE;
    __shared__ u32 wtile[BLOCK / WAVE][MK_ITERS];
    __shared__ u64 s_prefix;

    // phase 1: per-(wave, tile) counts
    u8 rel[VBYTES];
    for (int it = 0; it < MK_ITERS; ++it) {
        const long off = off0 + (long)it * TILE + (long)tid * VBYTES;
        u32 cnt = gather_marks(text, n, mode, off, rel);
        u32 tot = cnt;
        for (int d = WAVE / 2; d > 0; d >>= 1)
            tot += __shfl_down(tot, d, WAVE);
        if (lane == 0) wtile[wid][it] = tot;
    }
    __syncthreads();
    u32 agg = 0;
    for (int w = 0; w < BLOCK / WAVE; ++w)
        for (int it = 0; it < MK_ITERS; ++it) agg += wtile[w][it];

    // publish aggregate; wave-parallel decoupled lookback (64 status
    // words per probe) resolves the exclusive prefix
    if (wid == 0) {
        if (b == 0) {
            if (lane == 0) {
                __atomic_store_n((u64*)&status[0],
                                 MK_FLAG_INC | (u64)agg,
                                 __ATOMIC_RELEASE);
                s_prefix = 0;
            }
        } else {
            if (lane == 0)
                __atomic_store_n((u64*)&status[b], MK_FLAG_AGG | (u64)agg,
                                 __ATOMIC_RELEASE);
            u64 prefix = 0;
            long base = b;             // window is [base-64, base)
            while (true) {
                const long j = base - WAVE + lane;
                u64 st = 0;
                bool ready = false;
                while (!ready) {
                    st = (j >= 0)
                        ? __atomic_load_n((u64*)&status[j],
                                          __ATOMIC_ACQUIRE)
                        : MK_FLAG_INC;
                    ready = (__ballot((st >> 62) != 0) == ~0ULL);
                }
                const u64 inc_m = __ballot((st & MK_FLAG_INC) != 0
                                           && j >= 0);
                int cut = -1;
                if (inc_m) cut = 63 - __clzll(inc_m);
                u64 contrib = (j >= 0 && lane >= cut) ? MK_VAL(st) : 0;
                for (int d = WAVE / 2; d > 0; d >>= 1)
                    contrib += __shfl_down(contrib, d, WAVE);
                prefix += __shfl(contrib, 0, WAVE);
                if (inc_m || base - WAVE <= 0)
                    break;
                base -= WAVE;
            }
            if (lane == 0) {
                __atomic_store_n((u64*)&status[b],
                                 MK_FLAG_INC | (u64)(prefix + agg),
                                 __ATOMIC_RELEASE);
                s_prefix = prefix;
            }
        }
        if (lane == 0 && off0 + (long)MK_ITERS * TILE >= n)
            *total = s_prefix + agg;
    }
    __syncthreads();

    // phase 2: rescan (L2-hot) and scatter at resolved offsets
    u32 run = (u32)s_prefix;
    for (int it = 0; it < MK_ITERS; ++it) {
        u32 wb = run;
        u32 tile_tot = 0;
        for (int w = 0; w < BLOCK / WAVE; ++w) {
            if (w < wid) wb += wtile[w][it];
            tile_tot += wtile[w][it];
        }
        const long off = off0 + (long)it * TILE + (long)tid * VBYTES;
        const u32 cnt = gather_marks(text, n, mode, off, rel);
        u32 scan = cnt;
        for (int d = 1; d < WAVE; d <<= 1) {
            u32 x = __shfl_up(scan, d, WAVE);
            if (lane >= d) scan += x;
        }
        const u32 excl = wb + scan - cnt;
        for (u32 j = 0; j < cnt; ++j)
            out[excl + j] = (u32)(off + rel[j]);
        run += tile_tot;
    }
}

std::vector<torch::Tensor> marks(torch::Tensor text, long mode,
                                 torch::Tensor out_buf,
                                 torch::Tensor status_buf,
                                 torch::Tensor total_buf) {
    check_u8(text);
    long n = text.numel();
    long nblocks = (n + (long)MK_ITERS * TILE - 1)
                   / ((long)MK_ITERS * TILE);
    TORCH_CHECK(status_buf.numel() >= std::max(nblocks, 1L),
                "status buffer too small");
    if (n == 0) {
        total_buf.zero_();
        return {out_buf, total_buf};
    }
    hipLaunchKernelGGL(marks_lookback_kernel, dim3((u32)nblocks),
                       dim3(BLOCK), 0, cur_stream(),
                       text.data_ptr<u8>(), n, (int)mode,
                       (volatile u64*)status_buf.data_ptr(),
                       (u32*)out_buf.data_ptr(),
                       (u64*)total_buf.data_ptr());
    return {out_buf, total_buf};
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
std::vector<torch::Tensor> hj_emit(torch::Tensor keys_l,
                                   torch::Tensor t_keys,
                                   torch::Tensor t_head,
                                   torch::Tensor next,
                                   torch::Tensor offsets, long total,
                                   long left_outer, long track_matched,
                                   long nr);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rs_hist", &rs_hist, "radix pass histogram (bin-major)");
    m.def("rs_scatter", &rs_scatter, "stable radix scatter pass");
    m.def("seg_reduce", &seg_reduce,
          "segmented reduce over sorted runs (op 0=sum,1=min,2=max)");
    m.def("hj_build", &hj_build, "hash-join build (chained)");
    m.def("hj_count", &hj_count, "hash-join probe match counts");
    m.def("hj_emit", &hj_emit, "hash-join emit (l,r) row-index pairs");
    m.def("tsv_sizes", &tsv_sizes, "per-row TSV byte sizes");
    m.def("tsv_format", &tsv_format, "format token/df/idf rows as TSV");
    m.def("marks", &marks,
          "single-pass ordered mark positions + total (decoupled "
          "lookback; mode 0=newline, 1=token start)");
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
