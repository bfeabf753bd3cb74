"""Engine comparison on the reference's own benchmark workload (TF-IDF
document frequency; /root/reference/benchmarks/tf-idf-dampr.py is the
spec): the pure-Python reference vs the dampr_amd host engine vs the
device engine.

Usage:
    python benchmarks/compare_engines.py [--mb 100] [--skip-reference]

The reference run needs /root/reference on sys.path (it is executed, not
copied); skip it elsewhere with --skip-reference.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import numpy as np


def make_corpus(mb):
    from dampr_amd.gpu.corpus import synth_corpus
    return synth_corpus(mb << 20, vocab=100_000, seed=7)


def write_corpus(arr, path):
    with open(path, "wb") as fh:
        fh.write(arr.tobytes())


def tokenize(line):
    import re
    return set(re.split(r"[^\w]+", line.lower()))


def run_reference(path):
    sys.path.insert(0, "/root/reference")
    from dampr import Dampr as RefDampr
    t0 = time.perf_counter()
    docs = RefDampr.text(path)
    idfs = docs.flat_map(lambda line: tokenize(line)).count()
    results = idfs.run()
    n = sum(1 for _ in results)
    dt = time.perf_counter() - t0
    results.delete()
    sys.path.pop(0)
    return dt, n


def run_host_engine(path):
    from dampr_amd import Dampr
    t0 = time.perf_counter()
    docs = Dampr.text(path)
    idfs = docs.flat_map(lambda line: tokenize(line)).count()
    results = idfs.run()
    n = sum(1 for _ in results)
    dt = time.perf_counter() - t0
    results.delete()
    return dt, n


def run_device_engine(arr):
    import torch
    from dampr_amd.gpu.tfidf import TfidfEngine
    dev = torch.device("cuda:0")
    text = torch.from_numpy(arr).to(dev)
    eng = TfidfEngine(dev)
    eng.reset()
    eng.count_chunk(text)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.reset()
    eng.count_chunk(text)
    keys, df = eng.extract()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return dt, keys.numel()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=100)
    ap.add_argument("--skip-reference", action="store_true")
    args = ap.parse_args()

    arr = make_corpus(args.mb)
    path = "/tmp/dampr_bench_corpus.txt"
    write_corpus(arr, path)
    docs = int((arr == ord("\n")).sum())
    gb = args.mb / 1024.0

    def report(tag, dt, n):
        print("{:<28} {:8.2f} s  {:10.0f} rows/s  {:6.3f} GB/s  "
              "({} keys)".format(tag, dt, docs / dt, gb / dt, n))

    if not args.skip_reference and os.path.isdir("/root/reference"):
        report("reference (pure Python)", *run_reference(path))
    report("dampr_amd host engine", *run_host_engine(path))
    try:
        import torch
        if torch.cuda.is_available():
            report("dampr_amd device engine", *run_device_engine(arr))
    except ImportError:
        pass


if __name__ == "__main__":
    main()
