"""Multi-output run with a shared checkpointed subgraph: four statistics
computed over one tokenized input that executes once (Dampr.run merges the
DAGs and dedupes shared stages).

Usage: python examples/word_stats.py <file-or-dir-or-glob>

Role parity with the reference's examples/word-stats.py.
"""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import logging

from dampr_amd import Dampr


def main(fname):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(message)s")

    words = Dampr.text(fname, 1024 ** 2) \
        .flat_map(lambda line: line.split())

    top_words = words.count() \
        .sort_by(lambda wc: -wc[1])

    total = top_words.fold_by(lambda _wc: 1,
                              value=lambda wc: wc[1],
                              binop=lambda x, y: x + y)

    lengths = top_words.fold_by(lambda wc: len(wc[0]),
                                value=lambda wc: wc[1],
                                binop=lambda x, y: x + y) \
        .sort_by(lambda cl: cl[0])

    avg_len = lengths.map(lambda cl: cl[0] * cl[1]) \
        .a_group_by(lambda _x: 1).sum() \
        .join(total) \
        .reduce(lambda weighted, tot:
                next(weighted)[1] / float(next(tot)[1]))

    tc, tw, wl, awl = Dampr.run(total, top_words, lengths, avg_len,
                                name="word-stats")

    print("Total words:", tc.read(1)[0][1])
    print("\nTop 10 words")
    for word, count in tw.read(10):
        print(" ", word, count)
    print("\nLength histogram")
    for length, count in wl.read(20):
        print(" ", length, count)
    print("\nAverage word length:", awl.read(1)[0][1])


if __name__ == "__main__":
    main(sys.argv[1])
