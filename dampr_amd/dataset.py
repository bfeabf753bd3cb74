"""Storage engine: dataset readers and spill writers.

Role parity with the reference's ``dampr/dataset.py`` (reference:
dataset.py:43-629) with a different design:

* Spilled runs use a framed format — ``[u32 length | payload]`` where the
  payload is a pickled list of ``(key, value)`` records, zlib-compressed when
  ``settings.compress_level > 0`` (the reference nests batched pickle inside
  gzip files, dataset.py:119-159).
* One writer hierarchy: ``Spiller`` (sorted run generation, K3),
  ``PartitionedSpiller`` (map-side shuffle write, K2 — sorts the whole buffer
  once, then buckets, so every partition run is key-sorted by construction),
  ``CombineSpiller`` (map-side associative combine, K6),
  ``ContiguousWriter`` (unsorted reduce output) and ``SinkWriter`` (text part
  files).  The reference's dead/broken writers (``UnorderedWriter`` et al.,
  SURVEY.md §2.5) have no analog.
* ``MergeDataset`` is the k-way merge by key (K4), ``Dataset.grouped_read``
  the group segmentation (K5).  On the GPU these roles are HIP kernels; this
  module is the host/out-of-core tier.
"""
import heapq
import itertools
import os
import pickle
import struct
import zlib
from operator import itemgetter

from . import settings
from .memory import MemoryGovernor

_FRAME_HDR = struct.Struct("<I")


# --------------------------------------------------------------------------
# Readers
# --------------------------------------------------------------------------

class Chunker(object):
    """Lazily yields Datasets to process as independent map chunks."""

    def chunks(self):
        raise NotImplementedError()


class Dataset(Chunker):
    """A readable collection of (key, value) records.

    A Dataset is its own single chunk, so datasets and chunkers interchange
    anywhere an input is expected (reference: dataset.py:424-442).
    """

    def read(self):
        raise NotImplementedError()

    def grouped_read(self):
        """Yield (key, values) over consecutive equal keys.

        Groups are materialized so callers may hold or skip group iterators
        freely (join operators rely on this).  Only meaningful on key-sorted
        datasets (runs, merges).
        """
        for k, kvs in itertools.groupby(self.read(), key=itemgetter(0)):
            group = [kv[1] for kv in kvs]
            yield k, iter(group)

    def delete(self):
        pass

    def __iter__(self):
        return self.read()

    def chunks(self):
        yield self


class EmptyDataset(Dataset):
    def read(self):
        return iter(())


class MemoryDataset(Dataset):
    def __init__(self, kvs):
        self.kvs = kvs

    def read(self):
        return iter(self.kvs)

    def delete(self):
        self.kvs = []


class StreamDataset(Dataset):
    """Wraps a one-shot iterator of (k, v)."""

    def __init__(self, it):
        self.it = it

    def read(self):
        return self.it


class RunDataset(Dataset):
    """A spilled run file in the framed format written by RunWriter."""

    def __init__(self, path, compressed):
        self.path = path
        self.compressed = compressed

    def read(self):
        with open(self.path, "rb") as fh:
            while True:
                hdr = fh.read(_FRAME_HDR.size)
                if len(hdr) < _FRAME_HDR.size:
                    return
                (length,) = _FRAME_HDR.unpack(hdr)
                payload = fh.read(length)
                if self.compressed:
                    payload = zlib.decompress(payload)
                for kv in pickle.loads(payload):
                    yield kv

    def delete(self):
        try:
            os.unlink(self.path)
        except OSError:
            pass


class TextLineDataset(Dataset):
    """Newline-delimited text over a byte range of a file.

    Byte-range semantics follow the reference (dataset.py:452-482): a chunk
    that does not start at offset 0 skips its first partial line (owned by
    the previous chunk) and reads through the end of the line containing its
    last byte.  Keys are byte offsets of each line start.
    """

    def __init__(self, path, start=0, end=None):
        self.path = path
        self.start = start
        self.end = end

    def read(self):
        with open(self.path, "rb") as fh:
            size = os.fstat(fh.fileno()).st_size
            end = size if self.end is None else min(self.end, size)
            pos = self.start
            if pos > 0:
                # A line is owned by the chunk its first byte falls in; if
                # the previous byte is not a newline we are mid-line and the
                # previous chunk owns it.
                fh.seek(pos - 1)
                if fh.read(1) != b"\n":
                    skipped = fh.readline()
                    pos += len(skipped)
            else:
                fh.seek(0)
            while pos < end:
                line = fh.readline()
                if not line:
                    return
                yield pos, line.rstrip(b"\n").decode("utf-8", "replace")
                pos += len(line)

    def delete(self):
        try:
            os.unlink(self.path)
        except OSError:
            pass


class GzipLineDataset(Dataset):
    """Whole-file gzip text (not splittable)."""

    def __init__(self, path):
        self.path = path

    def read(self):
        import gzip
        pos = 0
        with gzip.open(self.path, "rb") as fh:
            for line in fh:
                yield pos, line.rstrip(b"\n").decode("utf-8", "replace")
                pos += len(line)

    def delete(self):
        try:
            os.unlink(self.path)
        except OSError:
            pass


class CatDataset(Dataset):
    """Concatenation of several datasets, in order.  Each member is its own
    map chunk (so ``read_input(a, b)`` parallelizes across taps)."""

    def __init__(self, datasets):
        self.datasets = list(datasets)

    def read(self):
        for ds in self.datasets:
            for kv in ds.read():
                yield kv

    def chunks(self):
        for ds in self.datasets:
            for chunk in ds.chunks():
                yield chunk

    def delete(self):
        for ds in self.datasets:
            ds.delete()


class MergeDataset(Dataset):
    """K-way merge of key-sorted datasets (K4)."""

    def __init__(self, datasets):
        self.datasets = list(datasets)

    def read(self):
        if len(self.datasets) == 1:
            return self.datasets[0].read()
        return heapq.merge(*(d.read() for d in self.datasets),
                           key=itemgetter(0))

    def delete(self):
        for ds in self.datasets:
            ds.delete()


class DMChunker(Chunker):
    """Adapts a stage's data mapping {partition: [datasets]} to map chunks."""

    def __init__(self, data_mapping):
        self.dm = data_mapping

    def chunks(self):
        for part in sorted(self.dm):
            for ds in self.dm[part]:
                yield ds


def merge_datasets(datasets):
    """Single dataset view over a list of sorted runs."""
    if len(datasets) > 1:
        return MergeDataset(datasets)
    if len(datasets) == 1:
        return datasets[0]
    return EmptyDataset()


def cat_datasets(datasets):
    if isinstance(datasets, Chunker):
        datasets = list(datasets.chunks())
    if len(datasets) > 1:
        return CatDataset(datasets)
    if len(datasets) == 1:
        return datasets[0]
    return EmptyDataset()


# --------------------------------------------------------------------------
# Writers
# --------------------------------------------------------------------------

class RunWriter(object):
    """Streams framed record batches to one run file."""

    def __init__(self, path):
        self.path = path
        self.compressed = settings.compress_level > 0
        self._fh = open(path, "wb", buffering=1 << 20)
        self._n = 0

    def write_records(self, kvs):
        bs = settings.batch_size
        for i in range(0, len(kvs), bs):
            payload = pickle.dumps(kvs[i:i + bs], pickle.HIGHEST_PROTOCOL)
            if self.compressed:
                payload = zlib.compress(payload, settings.compress_level)
            self._fh.write(_FRAME_HDR.pack(len(payload)))
            self._fh.write(payload)
        self._n += len(kvs)

    def close(self):
        self._fh.close()
        return RunDataset(self.path, self.compressed)


class DatasetWriter(object):
    """Writer interface: start / add_record / finished -> data mapping."""

    def start(self):
        pass

    def add_record(self, key, value):
        raise NotImplementedError()

    def finished(self):
        """Returns {partition: [Dataset, ...]}."""
        raise NotImplementedError()


class Spiller(DatasetWriter):
    """Sorted run generation (K3): buffer under the RSS watermark, then
    sort-by-key and emit one run per spill."""

    def __init__(self, fs, memory=False, governor=None, sort=True):
        self.fs = fs
        self.memory = memory
        self.sort = sort
        self.governor = governor or MemoryGovernor()
        self.buf = []
        self.runs = []

    def add_record(self, key, value):
        self.buf.append((key, value))
        if not self.memory and self.governor.over_watermark():
            self.flush()

    def add_records(self, it):
        """Bulk add: batches of records with one watermark check per
        batch — the per-record method-call overhead dominates pure-Python
        hot loops otherwise."""
        from itertools import islice
        while True:
            batch = list(islice(it, 16384))
            if not batch:
                return
            self.buf.extend(batch)
            if not self.memory and \
                    self.governor.over_watermark_bulk(len(batch)):
                self.flush()

    def flush(self):
        if not self.buf:
            return
        if self.sort:
            self.buf.sort(key=itemgetter(0))
        if self.memory:
            self.runs.append(MemoryDataset(self.buf))
        else:
            w = RunWriter(self.fs.get_file())
            w.write_records(self.buf)
            self.runs.append(w.close())
        self.buf = []
        self.governor.reset()

    def finished(self):
        self.flush()
        runs, self.runs = self.runs, []
        return {0: runs}


class PartitionedSpiller(DatasetWriter):
    """Map-side shuffle write (K2): one buffer, sorted once per spill, then
    bucketed by ``partition(key)`` — every partition run is key-sorted."""

    def __init__(self, fs, splitter, n_partitions, memory=False,
                 governor=None):
        self.fs = fs
        self.splitter = splitter
        self.n_partitions = n_partitions
        self.memory = memory
        self.governor = governor or MemoryGovernor()
        self.buf = []
        self.parts = {p: [] for p in range(n_partitions)}

    def add_record(self, key, value):
        self.buf.append((key, value))
        if not self.memory and self.governor.over_watermark():
            self.flush()

    def add_records(self, it):
        from itertools import islice
        while True:
            batch = list(islice(it, 16384))
            if not batch:
                return
            self.buf.extend(batch)
            if not self.memory and \
                    self.governor.over_watermark_bulk(len(batch)):
                self.flush()

    def flush(self):
        if not self.buf:
            return
        self.buf.sort(key=itemgetter(0))
        buckets = {}
        part = self.splitter.partition
        n = self.n_partitions
        for kv in self.buf:
            buckets.setdefault(part(kv[0], n), []).append(kv)
        for p, kvs in buckets.items():
            if self.memory:
                self.parts[p].append(MemoryDataset(kvs))
            else:
                w = RunWriter(self.fs.get_substage(
                    "p{}".format(p)).get_file())
                w.write_records(kvs)
                self.parts[p].append(w.close())
        self.buf = []
        self.governor.reset()

    def finished(self):
        self.flush()
        parts, self.parts = self.parts, {p: [] for p in
                                         range(self.n_partitions)}
        return parts


class CombineSpiller(DatasetWriter):
    """Map-side associative combine (K6): dict upsert with a key-count cap
    (the reference documents ``reduce_buffer`` but never reads it,
    SURVEY.md §2.5 — here it is real) plus the RSS watermark."""

    def __init__(self, fs, binop, memory=False, governor=None,
                 max_keys=None):
        self.fs = fs
        self.binop = binop
        self.memory = memory
        self.governor = governor or MemoryGovernor()
        self.max_keys = max_keys if max_keys and max_keys > 0 else None
        self.table = {}
        self.runs = []

    def add_record(self, key, value):
        t = self.table
        if key in t:
            t[key] = self.binop(t[key], value)
        else:
            t[key] = value
            if self.max_keys is not None and len(t) >= self.max_keys:
                self.flush()
                return
        if not self.memory and self.governor.over_watermark():
            self.flush()

    def add_records(self, it):
        """Bulk add: tight upsert loop over record batches, one
        watermark / max-keys check per batch."""
        from itertools import islice
        binop = self.binop
        while True:
            batch = list(islice(it, 16384))
            if not batch:
                return
            t = self.table
            for key, value in batch:
                if key in t:
                    t[key] = binop(t[key], value)
                else:
                    t[key] = value
            if self.max_keys is not None and len(t) >= self.max_keys:
                self.flush()
            elif not self.memory and \
                    self.governor.over_watermark_bulk(len(batch)):
                self.flush()

    def flush(self):
        if not self.table:
            return
        kvs = sorted(self.table.items(), key=itemgetter(0))
        if self.memory:
            self.runs.append(MemoryDataset(kvs))
        else:
            w = RunWriter(self.fs.get_file())
            w.write_records(kvs)
            self.runs.append(w.close())
        self.table = {}
        self.governor.reset()

    def finished(self):
        self.flush()
        runs, self.runs = self.runs, []
        return {0: runs}


class ContiguousWriter(DatasetWriter):
    """Unsorted streaming output (reduce results)."""

    def __init__(self, fs, memory=False):
        self.fs = fs
        self.memory = memory
        self.buf = []
        self._writer = None
        self.governor = MemoryGovernor()

    def add_record(self, key, value):
        self.buf.append((key, value))
        if not self.memory and (len(self.buf) >= settings.batch_size
                                and self.governor.over_watermark()
                                or len(self.buf) >= 4 * settings.batch_size):
            self._drain()

    def _drain(self):
        if self._writer is None:
            self._writer = RunWriter(self.fs.get_file())
        self._writer.write_records(self.buf)
        self.buf = []
        self.governor.reset()

    def finished(self):
        if self.memory:
            kvs, self.buf = self.buf, []
            return {0: [MemoryDataset(kvs)]}
        if self._writer is None and not self.buf:
            return {0: []}
        self._drain()
        return {0: [self._writer.close()]}


class SinkWriter(DatasetWriter):
    """Durable text part files; writes the VALUE only, one per line,
    str-formatting any value (reference semantics: dataset.py:264-282
    ``print(value, file=f)``)."""

    def __init__(self, path, part_id):
        self.dir = path
        self.path = os.path.join(path, "part-{}".format(part_id))
        self._fh = None

    def start(self):
        os.makedirs(self.dir, exist_ok=True)
        self._fh = open(self.path, "w", buffering=1 << 20)

    def add_record(self, key, value):
        self._fh.write("{}\n".format(value))

    def finished(self):
        self._fh.close()
        return {0: [TextLineDataset(self.path)]}
