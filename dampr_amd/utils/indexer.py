"""Keyed file index over sqlite3 (role parity: reference
dampr/utils/indexer.py:35-125).

Builds a per-file ``(key, offset)`` index via a pipeline, then answers
union / intersect keyed retrievals by re-reading the files at the stored
line offsets.
"""
import logging
import os
import sqlite3

from ..dampr import Dampr
from ..inputs import read_paths

log = logging.getLogger("dampr_amd")


class Indexer(object):
    def __init__(self, path, suffix=".index"):
        self.path = path
        self.suffix = suffix

    def get_idx(self, path):
        dirname, base = os.path.split(path)
        return os.path.join(dirname, "." + base + self.suffix)

    def exists(self, path):
        return os.path.isfile(self.get_idx(path))

    def open_db(self, path, delete=False):
        idx = self.get_idx(path)
        if delete and os.path.isfile(idx):
            os.unlink(idx)
        return sqlite3.connect(idx)

    def create_db(self, path):
        db = self.open_db(path, True)
        db.cursor().execute(
            "CREATE TABLE key_index (key text, offset integer)")
        return db

    def build(self, key_f, force=False):
        """Index every file under ``path``: for each line, key_f(line)
        yields the keys stored at that line's byte offset."""
        paths = sorted(read_paths(self.path, False))

        def index_file(fname):
            log.debug("indexing %s", fname)
            db = self.create_db(fname)

            def records():
                offset = 0
                with open(fname, "r", encoding="utf-8") as fh:
                    for line in fh:
                        for key in key_f(line):
                            yield key, offset
                        offset += len(line.encode("utf-8"))

            cur = db.cursor()
            cur.executemany("INSERT INTO key_index values (?, ?)",
                            records())
            db.commit()
            cur.execute("create index key_idx on key_index (key)")
            db.commit()
            cur.execute("select count(*) from key_index")
            return cur.fetchone()[0]

        return Dampr.memory(paths) \
            .filter(lambda fname: force or not self.exists(fname)) \
            .map(index_file) \
            .fold_by(key=lambda x: 1, binop=lambda x, y: x + y) \
            .read(name="indexing")

    def _retrieve(self, query):
        paths = list(read_paths(self.path, False))

        def read_db(fname):
            db = self.open_db(fname)
            cur = db.cursor()
            cur.execute(query)
            with open(fname, "r", encoding="utf-8") as fh:
                for (offset,) in cur:
                    fh.seek(offset)
                    yield fh.readline()

        return Dampr.memory(paths).flat_map(read_db)

    def union(self, keys):
        """Lines whose index keys match ANY of ``keys``."""
        if not isinstance(keys, (list, tuple)):
            keys = [keys]
        query = ("select distinct offset from key_index where key in ({}) "
                 "order by offset asc").format(
                     ",".join('"{}"'.format(k) for k in keys))
        return self._retrieve(query)

    def intersect(self, keys, min_match=None):
        """Lines whose index keys match at least ``min_match`` of ``keys``
        (default: all)."""
        if not isinstance(keys, (list, tuple)):
            keys = [keys]
        if min_match is None:
            min_match = len(keys)
        if isinstance(min_match, float):
            min_match = int(min_match * len(keys))
        query = ("select offset from (select offset, count(*) as c "
                 "from key_index where key in ({}) group by offset) "
                 "where c >= {} order by offset asc").format(
                     ",".join('"{}"'.format(k) for k in keys), min_match)
        return self._retrieve(query)
