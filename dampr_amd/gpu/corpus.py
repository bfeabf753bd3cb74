"""Synthetic text corpus generator (for benches/tests: no network, so the
TF-IDF workload runs on generated Zipf-distributed text).

Layout: fixed 8-byte cells = 7 lowercase letters + separator; every
``words_per_line``-th separator is a newline, so lines are uniform length.
Word ids are Zipf-distributed over the vocabulary, which reproduces the
skewed document-frequency profile TF-IDF cares about.
"""
import numpy as np


def vocab_words(vocab):
    """(vocab, 7) u8 matrix: word i = base-26 encoding, 'a'..'z'."""
    ids = np.arange(vocab, dtype=np.int64)
    cols = []
    for _ in range(7):
        cols.append((ids % 26).astype(np.uint8) + ord("a"))
        ids //= 26
    return np.stack(cols[::-1], axis=1)


def synth_corpus(n_bytes, vocab=100_000, words_per_line=12, seed=0,
                 zipf_a=1.3):
    """Returns a u8 numpy array of newline-delimited text, ~n_bytes long."""
    cell = 8
    line_bytes = words_per_line * cell
    n_lines = max(1, int(n_bytes) // line_bytes)
    n_tokens = n_lines * words_per_line
    rng = np.random.default_rng(seed)
    ids = (rng.zipf(zipf_a, size=n_tokens) - 1) % vocab
    table = np.concatenate(
        [vocab_words(vocab),
         np.full((vocab, 1), ord(" "), dtype=np.uint8)], axis=1)
    out = table[ids].reshape(n_lines, line_bytes)
    out[:, -1] = ord("\n")
    return out.reshape(-1)


def synth_corpus_device(n_bytes, device, vocab=100_000, words_per_line=12,
                        seed=0, zipf_a=1.3, chunk_lines=1 << 24):
    """Device-side corpus generator, same layout as ``synth_corpus``
    (8-byte cells, uniform lines).  Zipf ids come from the standard
    inverse-transform approximation z = floor(u^(-1/(a-1))) — the same
    heavy-tailed profile as numpy's rejection sampler, generated at HBM
    bandwidth so multi-GB bench corpora take seconds, not minutes, and
    the bench's timed region dominates its wall time.  Generates in
    chunks to bound transient f64 scratch.  Returns a u8 CUDA tensor."""
    import torch
    cell = 8
    line_bytes = words_per_line * cell
    n_lines = max(1, int(n_bytes) // line_bytes)
    table_np = np.concatenate(
        [vocab_words(vocab),
         np.full((vocab, 1), ord(" "), dtype=np.uint8)], axis=1)
    table = torch.from_numpy(table_np).to(device)
    out = torch.empty(n_lines * line_bytes, dtype=torch.uint8,
                      device=device)
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    inv = -1.0 / (zipf_a - 1.0)
    for lo in range(0, n_lines, chunk_lines):
        hi = min(lo + chunk_lines, n_lines)
        n_tok = (hi - lo) * words_per_line
        u = torch.rand(n_tok, generator=g, device=device,
                       dtype=torch.float64).clamp_min_(1e-12)
        z = u.pow_(inv).clamp_(max=2.0 ** 62).to(torch.int64)
        ids = (z - 1) % vocab
        chunk = table[ids].view(hi - lo, line_bytes)
        chunk[:, -1] = ord("\n")
        out[lo * line_bytes:hi * line_bytes] = chunk.view(-1)
    return out


def oracle_df(text_bytes):
    """Exact CPU oracle: {token: number of lines containing it} — the
    reference TF-IDF's doc-frequency semantics
    (benchmarks/tf-idf-dampr.py:9-14)."""
    import re
    from collections import Counter
    rx = re.compile(r"[^\w]+")
    counts = Counter()
    for line in bytes(text_bytes).decode("utf-8").split("\n"):
        toks = set(t for t in rx.split(line.lower()) if t)
        counts.update(toks)
    return dict(counts)
