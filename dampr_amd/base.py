"""Operator library: the objects executed inside stage workers.

Role parity with the reference's ``dampr/base.py``.  Notable deltas:

* ``Splitter`` uses the builtin salted hash like the reference (C speed;
  fork-stable within a run) — ``StableSplitter`` is the opt-in
  device-reproducible form over ``keyhash``.
* ``OuterJoin`` works — the reference's is dead code with an undefined
  variable and a wrong drain loop (reference: base.py:337-366, SURVEY.md
  §2.5).
* Join operators walk grouped streams the same sort-merge way (K8's CPU
  tier); on the GPU the same roles are HIP kernels.
"""
import os
import uuid

from .dataset import (CatDataset, EmptyDataset, MemoryDataset, MergeDataset,
                      RunWriter, StreamDataset, cat_datasets,
                      merge_datasets)
from .keyhash import partition_of


class Splitter(object):
    """Partition assignment for the host engine's shuffle.

    Uses the builtin ``hash`` (C speed): workers are *forked*, so the
    interpreter's per-process hash salt is shared and placement is
    consistent within a run — the same property the reference relies on
    (reference: base.py:6-8).  The device engine never sees these
    partitions; where CPU/GPU-stable keys are required (string dict,
    cross-rank exchange) the canonical ``keyhash`` functions are used
    instead.  ``StableSplitter`` is the opt-in device-reproducible form.
    """

    def partition(self, key, n_partitions):
        return hash(key) % n_partitions

    @staticmethod
    def check_start_method():
        """The builtin-hash placement is only consistent across workers
        because executor._ctx pins the *fork* start method (all workers
        share the parent's hash salt).  Called at executor pool start so
        a future spawn-based pool fails loudly here instead of silently
        scattering equal keys across partitions."""
        from .executor import _ctx
        assert _ctx.get_start_method() == "fork", (
            "Splitter's builtin-hash partitioning requires forked "
            "workers (shared hash salt); use StableSplitter with a "
            "spawn-based pool")


class StableSplitter(object):
    def partition(self, key, n_partitions):
        return partition_of(key, n_partitions)


# --------------------------------------------------------------------------
# Mappers
# --------------------------------------------------------------------------

class Mapper(object):
    """Consumes whole datasets; lowest-level map interface."""

    def map(self, *datasets):
        raise NotImplementedError()


class Streamable(object):
    """Consumes a (k, v) stream; fusable with other Streamables."""

    def stream(self, kvs):
        raise NotImplementedError()


class Map(Mapper, Streamable):
    """Standard record-at-a-time mapper around ``f(k, v) -> iter (k', v')``."""

    def __init__(self, mapper):
        assert not isinstance(mapper, Mapper)
        self.mapper = mapper

    def map(self, *datasets):
        assert len(datasets) == 1
        return self.stream(datasets[0].read())

    def stream(self, kvs):
        f = self.mapper
        for key, value in kvs:
            for nkv in f(key, value):
                yield nkv

    def __str__(self):
        return "Map[{}]".format(getattr(self.mapper, "__name__",
                                        type(self.mapper).__name__))

    __repr__ = __str__


class ComposedStreamable(Streamable):
    def __init__(self, left, right):
        assert isinstance(left, Streamable)
        assert isinstance(right, Streamable)
        self.left = left
        self.right = right

    def stream(self, kvs):
        return self.right.stream(self.left.stream(kvs))


class ComposedMapper(Mapper):
    def __init__(self, left, right):
        assert isinstance(left, Mapper)
        assert isinstance(right, Streamable)
        self.left = left
        self.right = right

    def map(self, *datasets):
        return self.right.stream(self.left.map(*datasets))


class BlockMapper(Mapper, Streamable):
    """User-extensible mapper with start/add/finish lifecycle."""

    def start(self):
        pass

    def add(self, key, value):
        raise NotImplementedError()

    def finish(self):
        return ()

    def map(self, *datasets):
        assert len(datasets) == 1
        return self.stream(datasets[0].read())

    def stream(self, kvs):
        self.start()
        for key, value in kvs:
            for out in self.add(key, value):
                yield out
        for out in self.finish():
            yield out


class StreamMapper(Mapper, Streamable):
    """Partition-level mapper around ``f(iter values) -> iter (k', v')``."""

    def __init__(self, streamer_f):
        self.streamer_f = streamer_f

    def map(self, *datasets):
        assert len(datasets) == 1
        return self.stream(datasets[0].read())

    def stream(self, kvs):
        return self.streamer_f(v for _k, v in kvs)

    def __str__(self):
        return "StreamMapper[{}]".format(
            getattr(self.streamer_f, "__name__",
                    type(self.streamer_f).__name__))

    __repr__ = __str__


class MapCrossJoin(Mapper):
    """Map-side nested-loop cross join (K9); optionally caches the right
    side in memory."""

    def __init__(self, crosser, cache):
        self.crosser = crosser
        self.cache = cache

    def map(self, *datasets):
        assert len(datasets) == 2
        left = cat_datasets(datasets[0])
        right = cat_datasets(datasets[1])
        if self.cache:
            cached = list(right.read())
            read_right = lambda: iter(cached)
        else:
            read_right = right.read
        for k1, v1 in left.read():
            for k2, v2 in read_right():
                for kv in self.crosser(k1, v1, k2, v2):
                    yield kv


class MapAllJoin(Mapper):
    """Map-side broadcast join: aggregates the whole right side once, then
    streams the left side against it."""

    def __init__(self, crosser, load_f=None):
        self.crosser = crosser
        self.load_f = load_f or (lambda d: [v for _k, v in d])

    def map(self, *datasets):
        assert len(datasets) == 2
        left = cat_datasets(datasets[0])
        right = self.load_f(cat_datasets(datasets[1]).read())
        for k, v in left.read():
            for kv in self.crosser(k, v, right):
                yield kv


# --------------------------------------------------------------------------
# Reducers
# --------------------------------------------------------------------------

class Reducer(object):
    def reduce(self, *datasets):
        raise NotImplementedError()

    def yield_groups(self, datasets):
        return merge_datasets(datasets).grouped_read()


class Reduce(Reducer):
    """Per-group reduce around ``f(key, iter values) -> value``."""

    def __init__(self, reducer):
        self.reducer = reducer

    def reduce(self, *datasets):
        assert len(datasets) == 1
        for k, vs in self.yield_groups(datasets[0]):
            yield k, self.reducer(k, vs)


class KeyedReduce(Reduce):
    """Reduce whose output value carries the key: (k, (k, result))."""

    def reduce(self, *datasets):
        for k, v in super(KeyedReduce, self).reduce(*datasets):
            yield k, (k, v)


class BlockReducer(Reducer):
    """User-extensible reducer with start/add/finish lifecycle."""

    def start(self):
        pass

    def add(self, key, values):
        raise NotImplementedError()

    def finish(self):
        return ()

    def reduce(self, *datasets):
        assert len(datasets) == 1
        self.start()
        for k, vs in self.yield_groups(datasets[0]):
            for out in self.add(k, vs):
                yield out
        for out in self.finish():
            yield out


class StreamReducer(Reducer):
    """Partition-level reducer around ``f(iter (k, iter vs)) -> iter kv``."""

    def __init__(self, stream_f):
        self.stream_f = stream_f

    def reduce(self, *datasets):
        assert len(datasets) == 1
        for nk, nv in self.stream_f(self.yield_groups(datasets[0])):
            yield nk, (nk, nv)

    def __str__(self):
        return "StreamReducer[{}]".format(
            getattr(self.stream_f, "__name__",
                    type(self.stream_f).__name__))

    __repr__ = __str__


def _advance(group_iter):
    return next(group_iter, None)


class InnerJoin(Reducer):
    """Sort-merge inner equi-join over two co-partitioned grouped streams
    (K8's CPU tier)."""

    def __init__(self, joiner_f, many=False):
        self.joiner_f = joiner_f
        self.many = many

    def reduce(self, *datasets):
        assert len(datasets) == 2
        g1 = self.yield_groups(datasets[0])
        g2 = self.yield_groups(datasets[1])
        left, right = _advance(g1), _advance(g2)
        while left is not None and right is not None:
            if left[0] < right[0]:
                left = _advance(g1)
            elif left[0] > right[0]:
                right = _advance(g2)
            else:
                k = left[0]
                out = self.joiner_f(k, left[1], right[1])
                if not self.many:
                    out = [out]
                for nv in out:
                    yield k, nv
                left, right = _advance(g1), _advance(g2)


class KeyedInnerJoin(InnerJoin):
    def reduce(self, *datasets):
        for k, v in super(KeyedInnerJoin, self).reduce(*datasets):
            yield k, (k, v)


class LeftJoin(Reducer):
    def __init__(self, joiner_f, default=lambda: iter(())):
        self.joiner_f = joiner_f
        self.default = default

    def reduce(self, *datasets):
        assert len(datasets) == 2
        g1 = self.yield_groups(datasets[0])
        g2 = self.yield_groups(datasets[1])
        left, right = _advance(g1), _advance(g2)
        while left is not None and right is not None:
            if left[0] < right[0]:
                yield left[0], self.joiner_f(left[0], left[1],
                                             self.default())
                left = _advance(g1)
            elif left[0] > right[0]:
                right = _advance(g2)
            else:
                yield left[0], self.joiner_f(left[0], left[1], right[1])
                left, right = _advance(g1), _advance(g2)
        while left is not None:
            yield left[0], self.joiner_f(left[0], left[1], self.default())
            left = _advance(g1)


class KeyedLeftJoin(LeftJoin):
    def reduce(self, *datasets):
        for k, v in super(KeyedLeftJoin, self).reduce(*datasets):
            yield k, (k, v)


class OuterJoin(Reducer):
    """Full outer sort-merge join.  The reference ships a broken, dead
    ``OuterJoin`` (base.py:337-366); this one works and is reachable via
    ``PJoin.outer_reduce``."""

    def __init__(self, joiner_f, default=lambda: iter(())):
        self.joiner_f = joiner_f
        self.default = default

    def reduce(self, *datasets):
        assert len(datasets) == 2
        g1 = self.yield_groups(datasets[0])
        g2 = self.yield_groups(datasets[1])
        left, right = _advance(g1), _advance(g2)
        while left is not None and right is not None:
            if left[0] < right[0]:
                yield left[0], self.joiner_f(left[0], left[1],
                                             self.default())
                left = _advance(g1)
            elif left[0] > right[0]:
                yield right[0], self.joiner_f(right[0], self.default(),
                                              right[1])
                right = _advance(g2)
            else:
                yield left[0], self.joiner_f(left[0], left[1], right[1])
                left, right = _advance(g1), _advance(g2)
        while left is not None:
            yield left[0], self.joiner_f(left[0], left[1], self.default())
            left = _advance(g1)
        while right is not None:
            yield right[0], self.joiner_f(right[0], self.default(),
                                          right[1])
            right = _advance(g2)


class KeyedOuterJoin(OuterJoin):
    def reduce(self, *datasets):
        for k, v in super(KeyedOuterJoin, self).reduce(*datasets):
            yield k, (k, v)


class CrossJoin(Reducer):
    """Reduce-side nested-loop cross of two partition streams."""

    def __init__(self, joiner_f):
        self.joiner_f = joiner_f

    def reduce(self, *datasets):
        assert len(datasets) == 2
        for lk, lv in merge_datasets(datasets[0]).read():
            for rk, rv in merge_datasets(datasets[1]).read():
                yield self.joiner_f(lk, lv, rk, rv)


class KeyedCrossJoin(CrossJoin):
    def reduce(self, *datasets):
        for k, v in super(KeyedCrossJoin, self).reduce(*datasets):
            yield k, (k, v)


# --------------------------------------------------------------------------
# Combiners and shuffler
# --------------------------------------------------------------------------

class Combiner(object):
    """Merges a worker's sorted runs into one stream before the shuffle."""

    def combine(self, datasets):
        raise NotImplementedError()


class NoopCombiner(Combiner):
    def combine(self, datasets):
        return MergeDataset(datasets)


class UnorderedCombiner(Combiner):
    def combine(self, datasets):
        return CatDataset(datasets)


class PartialReduceCombiner(Combiner):
    """Applies the associative reduce while merging sorted runs (K7's
    map-side half)."""

    def __init__(self, reducer):
        self.reducer = reducer

    def _fold(self, datasets):
        f = self.reducer.reducer
        for k, vs in MergeDataset(datasets).grouped_read():
            yield k, f(k, vs)

    def combine(self, datasets):
        return StreamDataset(self._fold(datasets))


class Shuffler(object):
    def __init__(self, n_partitions, splitter):
        self.n_partitions = n_partitions
        self.splitter = splitter

    def shuffle(self, fs, datasets):
        raise NotImplementedError()


class DefaultShuffler(Shuffler):
    """Streams a merged (key-sorted) run into per-partition contiguous runs;
    order is preserved, so each partition file stays key-sorted."""

    def __init__(self, n_partitions, splitter, memory=False):
        super(DefaultShuffler, self).__init__(n_partitions, splitter)
        self.memory = memory

    def shuffle(self, fs, datasets):
        part = self.splitter.partition
        n = self.n_partitions
        if self.memory:
            buckets = {p: [] for p in range(n)}
            for k, v in MergeDataset(datasets).read():
                buckets[part(k, n)].append((k, v))
            return {p: [MemoryDataset(kvs)] for p, kvs in buckets.items()}

        writers = {}
        bufs = {p: [] for p in range(n)}
        drained = 0
        for k, v in MergeDataset(datasets).read():
            p = part(k, n)
            bufs[p].append((k, v))
            drained += 1
            if drained >= 65536:
                for pp, kvs in bufs.items():
                    if kvs:
                        self._drain(writers, fs, pp, kvs)
                        bufs[pp] = []
                drained = 0
        out = {}
        for p in range(n):
            if bufs[p]:
                self._drain(writers, fs, p, bufs[p])
            if p in writers:
                out[p] = [writers[p].close()]
            else:
                out[p] = []
        return out

    def _drain(self, writers, fs, p, kvs):
        if p not in writers:
            writers[p] = RunWriter(
                fs.get_substage("shuf_{}".format(p)).get_file())
        writers[p].write_records(kvs)


# --------------------------------------------------------------------------
# Working-directory tree
# --------------------------------------------------------------------------

class _PathNode(object):
    def __init__(self, path):
        self.path = path

    def get_file(self, name=None):
        if name is None:
            name = uuid.uuid4().hex
        os.makedirs(self.path, exist_ok=True)
        return os.path.join(self.path, name)


class FileSystem(_PathNode):
    def get_stage(self, name):
        return StageFileSystem(os.path.join(self.path,
                                            "stage_{}".format(name)))


class StageFileSystem(_PathNode):
    def get_worker(self, w_id):
        return WorkerFileSystem(os.path.join(self.path,
                                             "worker_{}".format(w_id)))


class WorkerFileSystem(_PathNode):
    def get_substage(self, s):
        return SubStageFileSystem(os.path.join(self.path,
                                               "sub_{}".format(s)))


class SubStageFileSystem(_PathNode):
    pass
