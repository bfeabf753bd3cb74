"""Streaming TF-IDF ingest: corpora larger than any storage tier stream
through the engine chunk by chunk (generate -> H2D -> count -> discard);
only the count/dict tables stay resident.  This is the BASELINE config-5
(2 TB) mechanism at selectable scale.

Usage (GPU box):  python scripts/stream_tfidf.py [total_gb] [chunk_mb]
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np


def chunk_stream(total_bytes, chunk_bytes, vocab=100_000, seed0=1000):
    """Yields (chunk_np, pos_base): independent synthetic chunks, each
    newline-terminated so no line straddles a boundary."""
    from dampr_amd.gpu.corpus import synth_corpus
    pos = 0
    i = 0
    while pos < total_bytes:
        n = min(chunk_bytes, total_bytes - pos)
        chunk = synth_corpus(n, vocab=vocab, seed=seed0 + i)
        yield chunk, pos
        pos += chunk.nbytes
        i += 1


def main(total_gb=8, chunk_mb=1024, template=False):
    """``template=True`` pre-generates ONE chunk and re-streams it (the
    ingest path is identical; token distribution repeats per chunk — the
    honest mode for pure-throughput runs where host-side synthesis would
    otherwise dominate the wall clock).  Correctness invariant: every
    token's df is exactly n_chunks x its single-chunk df."""
    import torch
    from dampr_amd.gpu.tfidf import TfidfEngine
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    eng = TfidfEngine(dev)
    eng.reset()
    total = total_gb << 30
    gpu_s = 0.0
    t_all = time.perf_counter()
    if template:
        from dampr_amd.gpu.corpus import synth_corpus
        base = synth_corpus(chunk_mb << 20, vocab=100_000, seed=7)
        host = torch.from_numpy(base).pin_memory()
        n_chunks = (total + host.numel() - 1) // host.numel()
        t0 = time.perf_counter()
        for i in range(n_chunks):
            text = host.to(dev, non_blocking=True)
            eng.count_chunk(text, pos_base=i * host.numel())
            del text
        torch.cuda.synchronize()
        gpu_s = time.perf_counter() - t0
        total = n_chunks * host.numel()
    else:
        for chunk, pos in chunk_stream(total, chunk_mb << 20):
            t0 = time.perf_counter()
            text = torch.from_numpy(chunk).to(dev, non_blocking=True)
            eng.count_chunk(text, pos_base=pos)
            del text
            torch.cuda.synchronize()
            gpu_s += time.perf_counter() - t0
    keys, df = eng.extract()
    wall = time.perf_counter() - t_all
    docs = eng.n_docs
    print("streamed {} GiB: {} docs, {} distinct tokens".format(
        total_gb, docs, keys.numel()))
    print("wall {:.1f} s ({:.1f} GB/s end-to-end incl. host synth); "
          "H2D+count {:.1f} s ({:.1f} GB/s GPU-side)".format(
              wall, total / wall / 1e9, gpu_s, total / gpu_s / 1e9))


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 8,
         int(sys.argv[2]) if len(sys.argv) > 2 else 1024,
         template="--template" in sys.argv)
