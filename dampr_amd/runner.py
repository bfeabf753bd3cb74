"""Logical plan (DAG) and the physical multi-process runner.

Role parity with the reference's ``dampr/runner.py`` + ``stagerunner.py``:
``Graph`` is an immutable copy-on-add builder whose stage list is already
topological (builders append post-order); ``MTRunner`` interprets it stage by
stage — map stages partition and spill sorted runs, reduce stages transpose
{partition -> [runs from every worker]} and merge-reduce each partition, and
a compaction pass caps file fan-in (reference: runner.py:174-374).

The stage workers live here too (the reference splits them into
stagerunner.py); they run inside forked processes via ``executor.run_stage``.
"""
import logging
import math
import os

from . import settings
from .base import (Combiner, DefaultShuffler, Mapper, NoopCombiner,
                   PartialReduceCombiner, Reducer, Shuffler, Splitter,
                   FileSystem)
from .dataset import (CombineSpiller, ContiguousWriter, Chunker, Dataset,
                      DMChunker, EmptyDataset, MergeDataset,
                      PartitionedSpiller, SinkWriter, Spiller,
                      cat_datasets, merge_datasets)
from .executor import StageWorker, run_stage

log = logging.getLogger("dampr_amd")


# --------------------------------------------------------------------------
# Plan nodes
# --------------------------------------------------------------------------

class Source(object):
    """Identity of a stage output; global counter gives graph-union dedupe
    (reference: runner.py:17-33)."""

    _CNT = 0

    def __init__(self, name):
        self.name = name
        self.cnt = Source._CNT
        Source._CNT += 1

    def __hash__(self):
        return hash(self.cnt)

    def __eq__(self, other):
        return isinstance(other, Source) and self.cnt == other.cnt

    def __str__(self):
        return "Source[`{}`]".format(self.name)

    __repr__ = __str__


class GMap(object):
    def __init__(self, output, inputs, mapper, combiner=None, shuffler=None,
                 options=None):
        self.output = output
        self.inputs = inputs
        self.mapper = mapper
        self.combiner = combiner
        self.shuffler = shuffler
        self.options = options or {}

    def __repr__(self):
        return "GMap[{}]".format(self.output.name)


class GReduce(object):
    def __init__(self, output, inputs, reducer, options=None):
        self.output = output
        self.inputs = inputs
        self.reducer = reducer
        self.options = options or {}

    def __repr__(self):
        return "GReduce[{}]".format(self.output.name)


class GSink(object):
    def __init__(self, output, inputs, mapper, path, options=None):
        self.output = output
        self.inputs = inputs
        self.mapper = mapper
        self.path = path
        self.options = options or {}

    def __repr__(self):
        return "GSink[path={}]".format(self.path)


class Graph(object):
    """Immutable logical DAG; every add returns (Source, new Graph)."""

    def __init__(self):
        self.inputs = {}
        self.stages = []

    def _copy(self):
        ng = Graph()
        ng.inputs.update(self.inputs)
        ng.stages.extend(self.stages)
        return ng

    def add_input(self, dataset):
        ng = self._copy()
        src = Source("Input:{}".format(len(self.inputs)))
        ng.inputs[src] = dataset
        return src, ng

    def add_mapper(self, inputs, mapper, combiner=None, shuffler=None,
                   name=None, options=None):
        assert isinstance(mapper, Mapper)
        assert combiner is None or isinstance(combiner, Combiner)
        assert shuffler is None or isinstance(shuffler, Shuffler)
        assert all(isinstance(i, Source) for i in inputs)
        name = name or "Map: {}"
        src = Source(name.format(len(self.stages)))
        ng = self._copy()
        ng.stages.append(GMap(src, inputs, mapper, combiner, shuffler,
                              options))
        return src, ng

    def add_reducer(self, inputs, reducer, name=None, options=None):
        assert isinstance(reducer, Reducer)
        assert all(isinstance(i, Source) for i in inputs)
        name = name or "Reduce: {}"
        src = Source(name.format(len(self.stages)))
        ng = self._copy()
        ng.stages.append(GReduce(src, inputs, reducer, options))
        return src, ng

    def add_sink(self, inputs, mapper, path, name=None, options=None):
        assert isinstance(mapper, Mapper)
        assert all(isinstance(i, Source) for i in inputs)
        name = name or "Sink: {}"
        src = Source(name.format(path))
        ng = self._copy()
        ng.stages.append(GSink(src, inputs, mapper, path, options))
        return src, ng

    def union(self, other):
        """Merge two graphs, deduping shared stages by Source identity
        (reference: runner.py:127-135) — shared subtrees execute once."""
        ng = self._copy()
        ng.inputs.update(other.inputs)
        seen = {s.output for s in ng.stages}
        for s in other.stages:
            if s.output not in seen:
                ng.stages.append(s)
                seen.add(s.output)
        return ng


# --------------------------------------------------------------------------
# Stage workers (run inside forked processes)
# --------------------------------------------------------------------------

class _MapWorker(StageWorker):
    """Plain map stage: map -> hash-partitioned sorted runs (K1+K2+K3)."""

    def __init__(self, mapper, n_partitions, memory):
        self.mapper = mapper
        self.n_partitions = n_partitions
        self.memory = memory

    def setup(self, w_id, stage_fs):
        fs = stage_fs.get_worker("map_{}".format(w_id))
        self.writer = PartitionedSpiller(fs, Splitter(), self.n_partitions,
                                         memory=self.memory)

    def process(self, job):
        _t_id, chunk, supplemental = job
        self.writer.add_records(self.mapper.map(chunk, *supplemental))

    def finish(self):
        return self.writer.finished()


class _CombineMapWorker(StageWorker):
    """Map stage with combiner/shuffler (a_group_by path): map ->
    associative combine dict (K6) -> merge runs (+partial reduce, K7) ->
    partition shuffle write (K2)."""

    def __init__(self, mapper, combiner, n_partitions, memory, binop,
                 max_keys):
        self.mapper = mapper
        self.combiner = combiner or NoopCombiner()
        self.n_partitions = n_partitions
        self.memory = memory
        self.binop = binop
        self.max_keys = max_keys

    def setup(self, w_id, stage_fs):
        self.fs = stage_fs.get_worker("map_{}".format(w_id))
        if callable(self.binop):
            self.writer = CombineSpiller(self.fs, self.binop,
                                         memory=self.memory,
                                         max_keys=self.max_keys)
        else:
            self.writer = Spiller(self.fs, memory=self.memory)

    def process(self, job):
        _t_id, chunk, supplemental = job
        self.writer.add_records(self.mapper.map(chunk, *supplemental))

    def finish(self):
        runs = self.writer.finished()[0]
        if len(runs) > 1:
            stream = self.combiner.combine(runs)
        elif len(runs) == 1:
            stream = runs[0]
        else:
            stream = EmptyDataset()
        shuffler = DefaultShuffler(self.n_partitions, Splitter(),
                                   memory=self.memory)
        out = shuffler.shuffle(self.fs, [stream])
        # the shuffle consumed the pre-shuffle combine runs fully:
        # remove their spill files (they would otherwise leak in /tmp)
        for r in runs:
            r.delete()
        return out


class _ReduceWorker(StageWorker):
    """Reduce stage: per partition, merge all workers' runs and reduce
    (K4+K5+K7/K8)."""

    def __init__(self, reducer, memory):
        self.reducer = reducer
        self.memory = memory

    def setup(self, w_id, stage_fs):
        self.fs = stage_fs.get_worker("red_{}".format(w_id))
        self.outputs = {}

    def process(self, job):
        # One output file per partition: each is key-sorted (merge of sorted
        # runs), so the final MergeDataset over partition files restores a
        # global key order regardless of the hash → partition mapping.
        p_id, dataset_lists = job
        writer = ContiguousWriter(self.fs, memory=self.memory)
        for k, v in self.reducer.reduce(*dataset_lists):
            writer.add_record(k, v)
        self.outputs[p_id] = writer.finished()[0]

    def finish(self):
        return self.outputs


class _CombineFilesWorker(StageWorker):
    """Compaction: merge run lists to cap file fan-in
    (reference: runner.py:293-320)."""

    def __init__(self, combiner, memory):
        self.combiner = combiner
        self.memory = memory
        self.outputs = []

    def setup(self, w_id, stage_fs):
        self.fs = stage_fs.get_worker("cmb_{}".format(w_id))

    def process(self, job):
        tag, datasets = job
        writer = ContiguousWriter(self.fs, memory=self.memory)
        for k, v in self.combiner.combine(datasets).read():
            writer.add_record(k, v)
        for d in datasets:
            d.delete()
        self.outputs.append((tag, writer.finished()[0]))

    def finish(self):
        return self.outputs


class _SinkWorker(StageWorker):
    """Sink stage: map then write text part files (durable, no cleanup)."""

    def __init__(self, mapper, path):
        self.mapper = mapper
        self.path = path
        self.files = []

    def setup(self, w_id, stage_fs):
        pass

    def process(self, job):
        t_id, chunk, supplemental = job
        writer = SinkWriter(self.path, t_id)
        writer.start()
        for k, v in self.mapper.map(chunk, *supplemental):
            writer.add_record(k, v)
        self.files.extend(writer.finished()[0])

    def finish(self):
        return {0: self.files}


# --------------------------------------------------------------------------
# Runner
# --------------------------------------------------------------------------

class RunnerBase(object):
    def __init__(self, name, graph, working_dir="/tmp"):
        self.file_system = FileSystem(os.path.join(working_dir, name))
        self.graph = graph

    def run(self, outputs, cleanup=True):
        from .utils.trace import get_trace, trace_stage
        get_trace().clear()
        data = dict(self.graph.inputs)
        to_delete = set()
        for stage_id, stage in enumerate(self.graph.stages):
            log.info("Stage %s/%s: %r", stage_id + 1,
                     len(self.graph.stages), stage)
            input_data = [data[i] for i in stage.inputs]
            cleanup_stage = True
            with trace_stage(repr(stage)):
                if isinstance(stage, GMap):
                    dm = self.run_map(stage_id, input_data, stage)
                elif isinstance(stage, GReduce):
                    dm = self.run_reducer(stage_id, input_data, stage)
                elif isinstance(stage, GSink):
                    dm = self.run_sink(stage_id, input_data, stage)
                    cleanup_stage = False
                else:
                    raise TypeError(
                        "unknown stage type: {!r}".format(stage))
            data[stage.output] = dm
            if cleanup_stage:
                to_delete.add(stage.output)

        rets = []
        for source in outputs:
            d = data[source]
            if isinstance(d, Dataset):
                cd = [d]
            elif isinstance(d, Chunker):
                cd = list(d.chunks())
            else:
                cd = [ds for p in sorted(d) for ds in d[p]]
            rets.append(cd)
            to_delete.discard(source)

        rets = self.format_outputs(rets)
        if cleanup:
            for sd in to_delete:
                dm = data[sd]
                if isinstance(dm, dict):
                    for ds_list in dm.values():
                        for ds in ds_list:
                            ds.delete()
            # prune now-empty stage/worker directories (kept outputs and
            # their parents survive; removal is best-effort bottom-up)
            root = self.file_system.path
            for dirpath, _dirs, _files in sorted(
                    os.walk(root), key=lambda w: -len(w[0])):
                try:
                    os.rmdir(dirpath)
                except OSError:
                    pass
        log.info("Finished")
        return rets

    @staticmethod
    def collapse(worker_results):
        out = {}
        for dm in worker_results:
            for p, datasets in dm.items():
                out.setdefault(p, []).extend(datasets)
        return out


class MTRunner(RunnerBase):
    """Multi-process runner: the default execution engine on CPU.  On a GPU
    node the same plan lowers to the device engine (dampr_amd.gpu) for
    recognized operator chains."""

    def __init__(self, name, graph, n_maps=None, n_reducers=None,
                 n_partitions=None, max_files_per_stage=None,
                 working_dir="/tmp"):
        super(MTRunner, self).__init__(name, graph, working_dir)
        self.n_maps = n_maps or settings.max_processes
        self.n_reducers = n_reducers or settings.max_processes
        self.n_partitions = n_partitions or settings.partitions
        self.max_files_per_stage = (max_files_per_stage
                                    or settings.max_files_per_stage)

    # -- map ---------------------------------------------------------------

    def _map_jobs(self, data_mappings, concat=False):
        if concat:
            # Concat stages iterate every input's chunks (PMap.concat).
            def all_chunks():
                i = 0
                for dm in data_mappings:
                    if not isinstance(dm, Chunker):
                        dm = DMChunker(dm)
                    for chunk in dm.chunks():
                        yield (i, chunk, [])
                        i += 1
            return all_chunks()
        iter_dm = data_mappings[0]
        if not isinstance(iter_dm, Chunker):
            iter_dm = DMChunker(iter_dm)
        supplementary = []
        for dm in data_mappings[1:]:
            if not isinstance(dm, Chunker):
                dm = DMChunker(dm)
            supplementary.append(list(dm.chunks()))
        return ((i, chunk, supplementary)
                for i, chunk in enumerate(iter_dm.chunks()))

    def run_map(self, stage_id, data_mappings, stage):
        jobs = self._map_jobs(data_mappings,
                              concat=stage.options.get("concat", False))
        stage_fs = self.file_system.get_stage(stage_id)
        opts = stage.options
        n_maps = opts.get("n_maps", self.n_maps)
        memory = opts.get("memory", False)
        n_partitions = opts.get("n_partitions", self.n_partitions)

        if stage.combiner is None and stage.shuffler is None:
            factory = lambda: _MapWorker(stage.mapper, n_partitions, memory)
        else:
            binop = opts.get("binop")
            # per-stage reduce_buffer, else the settings backstop (bounds
            # the combine dict between amortized RSS checks)
            max_keys = opts.get("reduce_buffer")
            if max_keys is None:
                max_keys = settings.reduce_buffer
            factory = lambda: _CombineMapWorker(
                stage.mapper, stage.combiner, n_partitions, memory, binop,
                max_keys)

        results = run_stage(factory, jobs, n_maps, stage_fs)
        collapsed = self.collapse(results)
        return self._compact(collapsed, stage.combiner, n_maps, stage_fs,
                             memory)

    # -- compaction --------------------------------------------------------

    def _chunk_tasks(self, tag, datasets):
        fan_in = min(self.max_files_per_stage, self.n_maps)
        group = min(int(math.ceil(len(datasets) / float(fan_in))),
                    self.max_files_per_stage)
        return [((tag, i), datasets[s:s + group])
                for i, s in enumerate(range(0, len(datasets), group))]

    def _compact(self, collapsed, combiner, n_procs, stage_fs, memory):
        while True:
            tasks = []
            for p, datasets in collapsed.items():
                if len(datasets) > self.max_files_per_stage:
                    tasks.extend(self._chunk_tasks(p, datasets))
            if not tasks:
                return collapsed
            c = combiner or NoopCombiner()
            factory = lambda: _CombineFilesWorker(c, memory)
            new_collapsed = {p: [] for p in collapsed}
            for worker_out in run_stage(factory, iter(tasks), n_procs,
                                        stage_fs):
                for (p, _i), datasets in worker_out:
                    new_collapsed[p].extend(datasets)
            collapsed = new_collapsed

    # -- reduce ------------------------------------------------------------

    def run_reducer(self, stage_id, data_mappings, stage):
        keys = sorted({p for dm in data_mappings for p in dm})
        jobs = ((p, [dm.get(p, []) for dm in data_mappings]) for p in keys)
        stage_fs = self.file_system.get_stage(stage_id)
        opts = stage.options
        n_reducers = opts.get("n_reducers", self.n_reducers)
        memory = opts.get("memory", False)
        factory = lambda: _ReduceWorker(stage.reducer, memory)
        results = run_stage(factory, jobs, n_reducers, stage_fs)
        return self.collapse(results)

    # -- sink --------------------------------------------------------------

    def run_sink(self, stage_id, data_mappings, stage):
        jobs = self._map_jobs(data_mappings)
        n_maps = stage.options.get("n_maps", self.n_maps)
        stage_fs = self.file_system.get_stage(stage_id)
        factory = lambda: _SinkWorker(stage.mapper, stage.path)
        results = run_stage(factory, jobs, n_maps, stage_fs)
        return self.collapse(results)

    # -- outputs -----------------------------------------------------------

    def format_outputs(self, outputs):
        rets = []
        for output in outputs:
            while len(output) > self.max_files_per_stage:
                log.debug("final combine over %d files", len(output))
                stage_fs = self.file_system.get_stage("final_combine")
                factory = lambda: _CombineFilesWorker(NoopCombiner(), False)
                jobs = self._chunk_tasks(None, output)
                output = [ds
                          for wout in run_stage(factory, iter(jobs),
                                                self.n_maps, stage_fs)
                          for _tag, datasets in wout
                          for ds in datasets]
            if len(output) == 1:
                rets.append(output[0])
            else:
                rets.append(MergeDataset(output))
        return rets
