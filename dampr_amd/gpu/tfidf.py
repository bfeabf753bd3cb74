"""Single-GPU TF-IDF pipeline over the device hash-combine engine.

This is the flagship benchmark path (BASELINE.json config 2): the
reference's ``flat_map(set(tokenize)).count().cross_right(docs.len(), idf)
.sink_tsv`` (benchmarks/tf-idf-dampr.py:9-21) re-built MI355X-native:

  text (resident in HBM)
    -> newline position scan                  (hand-written HIP, K1-adjacent)
    -> tokenize+hash+per-doc-dedupe+df-count  (one fused group-stream
       kernel: wave-wide segmented-XOR token hashing, K1+K6)
    -> [multi-GPU: RCCL all-to-all exchange of (key, df) partials]
    -> idf epilogue (K9's scalar broadcast-apply, fused elementwise)
    -> token-string gather + TSV sink

No sort is needed anywhere on the hot path — grouping is the device hash
table, exactly where the reference uses its in-memory combine dict
(dataset.py:84-117).
"""
import os

import torch

from ..ops import native

MODE_NEWLINE = 0
MODE_TOKEN_START = 1


def _pow2_at_least(n):
    return 1 << max(4, int(n - 1).bit_length())


class TfidfEngine(object):
    def __init__(self, device, vocab_capacity=1 << 21):
        self.ext = native.require()
        self.device = torch.device(device)
        self.cap = _pow2_at_least(vocab_capacity)
        opts = dict(dtype=torch.int64, device=self.device)
        self.cnt_keys = torch.zeros(self.cap, **opts)
        self.cnt_vals = torch.zeros(self.cap, **opts)
        self.dict_keys = torch.zeros(self.cap, **opts)
        self.dict_vals = torch.zeros(self.cap, **opts)
        self.seen = None
        self._fb_seen = None
        self._err = None
        self._ablate = 0        # perf-ablation bits; 0 = full work
        self.n_docs = 0

    def reset(self):
        self.cnt_keys.zero_()
        self.cnt_vals.zero_()
        self.dict_keys.zero_()
        self.dict_vals.zero_()
        self.n_docs = 0

    # -- scans ---------------------------------------------------------------

    def positions(self, text, mode):
        counts = self.ext.mark_counts(text, mode).to(torch.int64)
        offsets = torch.cumsum(counts, 0) - counts
        total = int(offsets[-1].item() + counts[-1].item())
        out = torch.empty(total, dtype=torch.int32, device=self.device)
        self.ext.mark_positions(text, mode, offsets.to(torch.int32), out)
        return out, total

    # -- one chunk -----------------------------------------------------------

    def count_chunk(self, text, pos_base=0):
        """text: u8 device tensor of newline-delimited ASCII.  ``pos_base``
        is the chunk's absolute byte offset in the job's corpus (dict
        entries store absolute positions so multi-chunk gathers work).

        Fast path: the group-stream kernel (uint4-staged LDS windows,
        ballot token starts, segmented-XOR hash scan, doc-salted LDS-set
        dedupe).  Documents whose distinct-token count overflows both the
        LDS set and the global fallback seen-table trip an error flag and
        the chunk reruns on the fully general token-centric kernel."""
        assert text.numel() < (1 << 31), "chunk must be < 2 GiB"
        nl, n_nl = self.positions(text, MODE_NEWLINE)
        # docs = newlines (+1 unterminated tail line)
        n = text.numel()
        tail = 0
        if n and int(text[-1].item()) != ord("\n"):
            tail = 1
        n_docs = n_nl + tail
        doc_base = self.n_docs
        self.n_docs += n_docs

        if self._fb_seen is None:
            self._fb_seen = torch.zeros(1 << 22, dtype=torch.int64,
                                        device=self.device)
            self._err = torch.zeros(2, dtype=torch.int32,
                                    device=self.device)
        else:
            self._fb_seen.zero_()
            self._err.zero_()
        # Snapshot the tables (~20 us for 64 MB at HBM rate) so an overflow
        # retry can roll back cleanly.
        snap = (self.cnt_keys.clone(), self.cnt_vals.clone(),
                self.dict_keys.clone(), self.dict_vals.clone())
        self.ext.tfidf_count_docs(text, nl, n_docs, self.cnt_keys,
                                  self.cnt_vals, self.dict_keys,
                                  self.dict_vals, pos_base,
                                  self._fb_seen, self._err, self._ablate)
        if int(self._err[0].item()):
            # Rerun this chunk on the general path with a full-size
            # (doc,token) seen table.
            (self.cnt_keys, self.cnt_vals,
             self.dict_keys, self.dict_vals) = snap
            self._count_chunk_general(text, nl, pos_base, doc_base)
        return n_docs

    def _count_chunk_general(self, text, nl, pos_base, doc_base):
        ts, n_tok = self.positions(text, MODE_TOKEN_START)
        seen_cap = _pow2_at_least(2 * max(n_tok, 1))
        if self.seen is None or self.seen.numel() < seen_cap:
            self.seen = torch.zeros(seen_cap, dtype=torch.int64,
                                    device=self.device)
        else:
            self.seen.zero_()
        self.ext.tfidf_count(text, nl, ts, self.seen, self.cnt_keys,
                             self.cnt_vals, self.dict_keys, self.dict_vals,
                             pos_base, doc_base)

    # -- table io ------------------------------------------------------------

    def extract(self):
        """(keys u64-as-i64, df i64), key-sorted for determinism."""
        n_out = int((self.cnt_keys != 0).sum().item())
        out_k, out_v, _cur = self.ext.table_extract(
            self.cnt_keys, self.cnt_vals, n_out)
        order = torch.argsort(out_k)
        return out_k[order], out_v[order]

    def merge_pairs(self, keys, vals):
        """Add (key, df) pairs (e.g. from the RCCL exchange) into the local
        table."""
        self.ext.table_merge(keys, vals, self.cnt_keys, self.cnt_vals)

    # -- epilogue ------------------------------------------------------------

    def idf(self, df, total_docs):
        return self.ext.idf(df, float(total_docs))

    def token_strings_dev(self, keys, text):
        """Device-side token materialization: (blob u8 dev, lens i64 dev)."""
        packed = self.ext.table_lookup(self.dict_keys, self.dict_vals, keys)
        lens = (packed & 0xFF).to(torch.int64)
        offsets = torch.cumsum(lens, 0) - lens
        total = int((offsets[-1] + lens[-1]).item()) if keys.numel() else 0
        blob = self.ext.gather_tokens(text, packed, offsets, total)
        return blob, lens

    def merge_exchanged(self, keys, df, blob, lens):
        """After the RCCL exchange: rebuild the owned shard's tables from
        received (key, df, token-bytes) partials."""
        self.cnt_keys.zero_()
        self.cnt_vals.zero_()
        self.dict_keys.zero_()
        self.dict_vals.zero_()
        self.ext.table_merge(keys, df, self.cnt_keys, self.cnt_vals)
        offsets = torch.cumsum(lens, 0) - lens
        packed = torch.bitwise_or(torch.bitwise_left_shift(offsets, 8),
                                  lens)
        self.ext.table_put(keys, packed, self.dict_keys, self.dict_vals)

    def token_strings(self, keys, text):
        """Materialize the token bytes for ``keys`` from the string dict.
        Returns (bytes_cpu, lengths_cpu)."""
        packed = self.ext.table_lookup(self.dict_keys, self.dict_vals, keys)
        lens = (packed & 0xFF).to(torch.int64)
        offsets = torch.cumsum(lens, 0) - lens
        total = int((offsets[-1] + lens[-1]).item()) if keys.numel() else 0
        blob = self.ext.gather_tokens(text, packed, offsets, total)
        return blob.cpu().numpy(), lens.cpu().numpy()

    def sink_tsv_device(self, path, part_id, keys, df, idf, src_text):
        """Device-formatted TSV sink: token\\tdf\\tidf rows are laid out by
        the tsv_format kernel; the host does one write()."""
        os.makedirs(path, exist_ok=True)
        packed = self.ext.table_lookup(self.dict_keys, self.dict_vals, keys)
        lens = (packed & 0xFF).to(torch.int64)
        tok_off = torch.cumsum(lens, 0) - lens
        sizes = self.ext.tsv_sizes(lens, df, idf)
        row_off = torch.cumsum(sizes, 0) - sizes
        if keys.numel() == 0:
            total_tok = 0
            total = 0
        else:
            total_tok = int((tok_off[-1] + lens[-1]).item())
            total = int((row_off[-1] + sizes[-1]).item())
        blob = self.ext.gather_tokens(src_text, packed, tok_off, total_tok)
        out = self.ext.tsv_format(blob, tok_off, lens, df, idf, row_off,
                                  total)
        with open(os.path.join(path, "part-{}".format(part_id)), "wb") as fh:
            fh.write(out.cpu().numpy().tobytes())

    def sink_tsv(self, path, part_id, tokens_blob, lens, df, idf):
        """Write the classic `token\\tdf\\tidf` part file."""
        os.makedirs(path, exist_ok=True)
        df_np = df.cpu().numpy()
        idf_np = idf.cpu().numpy()
        out = []
        pos = 0
        blob = tokens_blob.tobytes()
        for i in range(len(lens)):
            ln = int(lens[i])
            tok = blob[pos:pos + ln].decode("ascii")
            pos += ln
            out.append("{}\t{}\t{}".format(tok, df_np[i], idf_np[i]))
        with open(os.path.join(path, "part-{}".format(part_id)), "w") as fh:
            fh.write("\n".join(out))
            if out:
                fh.write("\n")


def run_tfidf(text_np, device="cuda:0", sink_path=None, chunk_bytes=None,
              engine=None):
    """One full single-GPU TF-IDF job over a host (numpy u8) or
    device-resident (torch u8) corpus; returns {token: (df, idf)}
    (also sinks TSV when sink_path given)."""
    dev = torch.device(device)
    if isinstance(text_np, torch.Tensor):
        text = text_np.to(dev)
        text_np = text.cpu().numpy()
    else:
        text = torch.from_numpy(text_np).to(dev)
    eng = engine or TfidfEngine(dev)
    eng.reset()
    n = text.numel()
    cb = chunk_bytes or n
    # Chunk on newline boundaries so no line straddles two chunks.
    import numpy as np
    bounds = [0]
    while bounds[-1] < n:
        e = min(bounds[-1] + cb, n)
        if e < n:
            nl = np.flatnonzero(text_np[e - 1:min(e + (1 << 16), n)]
                                == ord("\n"))
            e = (e - 1 + int(nl[0]) + 1) if len(nl) else n
        bounds.append(e)
    for s, e in zip(bounds, bounds[1:]):
        eng.count_chunk(text[s:e].contiguous(), pos_base=s)
    keys, df = eng.extract()
    idf = eng.idf(df, eng.n_docs)
    blob, lens = eng.token_strings(keys, text)
    if sink_path:
        eng.sink_tsv_device(sink_path, 0, keys, df, idf, text)
    out = {}
    pos = 0
    b = blob.tobytes()
    df_np = df.cpu().numpy()
    idf_np = idf.cpu().numpy()
    for i in range(len(lens)):
        ln = int(lens[i])
        out[b[pos:pos + ln].decode("ascii")] = (int(df_np[i]),
                                                float(idf_np[i]))
        pos += ln
    return out
