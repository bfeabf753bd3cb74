"""Word count over text files on the multi-process host engine.

Usage: python examples/wc.py <file-or-dir-or-glob>

Role parity with the reference's examples/wc.py; written against the
dampr_amd DSL.
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

    wc = Dampr.text(fname) \
        .flat_map(lambda line: line.split()) \
        .count() \
        .sort_by(lambda word_count: -word_count[1])

    results = wc.run("word-count")
    for word, count in results:
        print("{}: {}".format(word, count))
    results.delete()


if __name__ == "__main__":
    main(sys.argv[1])
