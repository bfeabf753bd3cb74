"""The flagship single-GPU TF-IDF pipeline (BASELINE config 2): tokenize +
per-document dedupe + document-frequency count + idf + TSV sink, all on
device (one fused pass over the text; see gpu/tfidf.py).

Usage (on an MI355X): python examples/tfidf_device.py [corpus-mb]
"""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))


import torch


def main(mb=64):
    if not torch.cuda.is_available():
        raise SystemExit("this example needs a GPU (see examples/wc.py "
                         "for the host engine)")
    from dampr_amd.gpu.corpus import synth_corpus
    from dampr_amd.gpu.tfidf import run_tfidf

    text = synth_corpus(mb << 20, vocab=50_000, seed=3)
    out = run_tfidf(text, device="cuda:0",
                    sink_path="/tmp/dampr_amd_example_idfs")
    top = sorted(out.items(), key=lambda kv: -kv[1][0])[:10]
    print("highest-df tokens:")
    for tok, (df, idf) in top:
        print("  {:<12} df={:<8} idf={:.4f}".format(tok, df, idf))
    print("TSV parts under /tmp/dampr_amd_example_idfs/")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 64)
