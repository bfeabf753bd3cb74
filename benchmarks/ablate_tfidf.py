"""Perf ablation of the doc-centric count kernel (run on a GPU box)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dampr_amd.gpu.corpus import synth_corpus
from dampr_amd.gpu.tfidf import TfidfEngine

def timed(eng, text, ablate, iters=3):
    eng._ablate = ablate
    # warmup
    eng.reset(); eng.count_chunk(text); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        eng.reset()
        eng.count_chunk(text)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

def main():
    dev = torch.device("cuda:0")
    text = torch.from_numpy(synth_corpus(1 << 30, vocab=100_000,
                                         seed=7)).to(dev)
    eng = TfidfEngine(dev)
    for name, ab in [("hash-only(stage+detect+walk)", 1),
                     ("+dedupe(LDS set)", 2),
                     ("+count(no dict)", 4),
                     ("full", 0)]:
        print("{:35s} {:8.2f} ms".format(name, timed(eng, text, ab)))

if __name__ == "__main__":
    main()
