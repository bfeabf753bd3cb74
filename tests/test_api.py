"""Conformance suite: black-box API tests through the real multi-process
engine, covering the reference's 36-test feature matrix plus
the additions (outer join, concat, fault detection).

Every test runs tiny data through the full fork/spill/shuffle machinery, the
same style the reference uses as its executable spec.
"""
import itertools
import os
import shutil

import pytest

from dampr_amd import Dampr, BlockMapper, BlockReducer, Dataset, settings
from dampr_amd.utils import filter_by_count


class RangeDataset(Dataset):
    def __init__(self, n):
        self.n = n

    def read(self):
        for i in range(self.n):
            yield i, i


@pytest.fixture
def items():
    return Dampr.memory(list(range(10, 20)), partitions=2)


def test_identity(items):
    assert list(items.run()) == list(range(10, 20))


def test_map(items):
    assert list(items.map(lambda x: x + 1).run()) == list(range(11, 21))


def test_group_by_count(items):
    res = items.group_by(lambda x: 1, lambda x: 1) \
               .reduce(lambda k, it: sum(it)).run()
    assert next(iter(res))[1] == 10


def test_count_with_none_key(items):
    res = items.count(lambda x: None).run()
    assert next(iter(res)) == (None, 10)


def test_group_by_sum(items):
    res = items.group_by(lambda x: 1).reduce(lambda k, it: sum(it)).run()
    assert next(iter(res))[1] == sum(range(10, 20))

    res = items.group_by(lambda v: v % 2) \
               .reduce(lambda k, it: sum(it)).run()
    assert [kv[1] for kv in res] == [10 + 12 + 14 + 16 + 18,
                                     11 + 13 + 15 + 17 + 19]


def test_filter(items):
    odds = items.filter(lambda i: i % 2 == 1).run()
    assert list(odds) == [11, 13, 15, 17, 19]


def test_sort_by(items):
    res = items.sort_by(lambda x: -x).run()
    assert list(res) == list(range(19, 9, -1))


def test_reduce_join(items):
    items2 = Dampr.memory(list(range(10)))
    res = items.group_by(lambda x: x % 2) \
        .join(items2.group_by(lambda x: x % 2)) \
        .reduce(lambda l, r: sorted(itertools.chain(l, r))) \
        .run()
    output = list(res)
    assert output[0] == (0, [0, 2, 4, 6, 8, 10, 12, 14, 16, 18])
    assert output[1] == (1, [1, 3, 5, 7, 9, 11, 13, 15, 17, 19])


def test_join_disjoint_keys(items):
    items2 = Dampr.memory(list(range(10))).group_by(lambda x: -x)
    output = items.group_by(lambda x: x).join(items2).run()
    assert [v for _k, v in output] == []


def test_join_after_reduce_repartition(items):
    items2 = Dampr.memory(list(range(10))) \
        .group_by(lambda x: -x).reduce(lambda k, vs: sum(vs))
    output = items.group_by(lambda x: x).join(items2).run()
    assert [v for _k, v in output] == []


def test_associative_reduce(items):
    output = list(items.a_group_by(lambda x: x % 2)
                  .reduce(lambda x, y: x + y).run())
    assert output[0][1] == 10 + 12 + 14 + 16 + 18
    assert output[1][1] == 11 + 13 + 15 + 17 + 19


def test_left_join(items):
    to_remove = Dampr.memory(list(range(10, 13)))
    output = items.group_by(lambda x: x) \
        .join(to_remove.group_by(lambda x: x)) \
        .left_reduce(lambda l, r: (list(l), list(r))) \
        .filter(lambda llrs: len(llrs[1][1]) == 0) \
        .map(lambda llrs: llrs[1][0][0]) \
        .sort_by(lambda x: x) \
        .run()
    assert list(output) == list(range(13, 20))


def test_outer_join(items):
    # New capability: the reference's outer join is broken dead code.
    left = Dampr.memory([1, 2, 3])
    right = Dampr.memory([2, 3, 4])
    output = left.group_by(lambda x: x) \
        .join(right.group_by(lambda x: x)) \
        .outer_reduce(lambda l, r: (list(l), list(r))) \
        .run()
    got = sorted((k, v) for k, v in output)
    assert got == [(1, ([1], [])), (2, ([2], [2])),
                   (3, ([3], [3])), (4, ([], [4]))]


def test_multi_output_run(items):
    even = items.filter(lambda x: x % 2 == 0)
    odd = items.filter(lambda x: x % 2 == 1)
    even_ve, odd_ve = Dampr.run(even, odd)
    assert list(even_ve) == [10, 12, 14, 16, 18]
    assert list(odd_ve) == [11, 13, 15, 17, 19]


def test_reduce_many(items):
    even = items.filter(lambda x: x % 2 == 0)
    odd = items.filter(lambda x: x % 2 == 1)

    def cross(x, y):
        y = list(y)
        for xi in x:
            for yi in y:
                yield xi * yi

    results = even.group_by(lambda x: 1) \
        .join(odd.group_by(lambda x: 1)) \
        .reduce(cross, many=True) \
        .run().read()
    expected = sorted((1, ei * oi)
                      for ei in [10, 12, 14, 16, 18]
                      for oi in [11, 13, 15, 17, 19])
    assert sorted(results) == expected


def test_fold_by(items):
    output = items.fold_by(lambda x: 1, value=lambda x: x % 2,
                           binop=lambda x, y: x + y)
    assert list(output.run()) == [(1, 5)]


def test_empty_map(items):
    output = items.sample(0.0).fold_by(lambda x: 1,
                                       value=lambda x: x % 2,
                                       binop=lambda x, y: x + y)
    assert list(output.run()) == []


def test_sink(items):
    path = "/tmp/dampr_amd_test_sink"
    shutil.rmtree(path, ignore_errors=True)
    sink = items.map(lambda x: str(x)).sink(path=path)
    output = sink.count()
    results = sorted(output.run())
    assert results == [(str(i), 1) for i in range(10, 20)]
    assert os.path.isdir(path)
    shutil.rmtree(path)


def test_sink_tsv(items):
    path = "/tmp/dampr_amd_test_sink_tsv"
    shutil.rmtree(path, ignore_errors=True)
    items.map(lambda x: (x, x * 2)).sink_tsv(path).run()
    lines = []
    for part in os.listdir(path):
        with open(os.path.join(path, part)) as fh:
            lines.extend(l.rstrip("\n") for l in fh)
    assert sorted(lines) == sorted("{}\t{}".format(i, i * 2)
                                   for i in range(10, 20))
    shutil.rmtree(path)


def test_sink_json(items):
    path = "/tmp/dampr_amd_test_sink_json"
    shutil.rmtree(path, ignore_errors=True)
    items.sink_json(path).run()
    import json
    vals = []
    for part in os.listdir(path):
        with open(os.path.join(path, part)) as fh:
            vals.extend(json.loads(l) for l in fh)
    assert sorted(vals) == list(range(10, 20))
    shutil.rmtree(path)


def test_cached(items):
    sink = items.map(lambda x: str(x)).cached()
    sink.run()
    results = sorted(sink.count().run())
    assert results == [(str(i), 1) for i in range(10, 20)]


def test_cross_right_scalar(items):
    total = items.a_group_by(lambda x: 1).sum()
    output = items.cross_right(
        total, lambda v1, v2: round(v1 / float(v2[1]), 4)) \
        .sort_by(lambda x: x)
    count = sum(range(10, 20))
    assert sorted(output.run()) == [round(i / float(count), 4)
                                    for i in range(10, 20)]


def test_cross_left_multi(items):
    output = items.cross_left(items, lambda v1, v2: v1 * v2)
    expected = sorted(i * k
                      for i in range(10, 20) for k in range(10, 20))
    assert sorted(output.run()) == expected


def test_cross_set(items):
    # Matches the reference's *actual* semantics (verified against it):
    # iterate `other`'s values against the aggregated self.
    right = Dampr.memory([13, 15, 99])
    output = items.cross_set(right, lambda x, y: x in y, agg=set)
    assert sorted(output.run(), key=str) == [False, True, True]


def test_custom_blocks():
    import heapq

    class TopKMapper(BlockMapper):
        def __init__(self, k):
            self.k = k

        def start(self):
            self.heap = []

        def add(self, _k, lc):
            heapq.heappush(self.heap, (lc[1], lc[0]))
            if len(self.heap) > self.k:
                heapq.heappop(self.heap)
            return iter(())

        def finish(self):
            for cl in self.heap:
                yield 1, cl

    class TopKReducer(BlockReducer):
        def __init__(self, k):
            self.k = k

        def add(self, k, it):
            for count, letter in heapq.nlargest(self.k, it):
                yield letter, (letter, count)

    word = Dampr.memory(["supercalifragilisticexpialidociousa"])
    letter_counts = word.flat_map(list).count()
    topk = letter_counts.custom_mapper(TopKMapper(2)) \
                        .custom_reducer(TopKReducer(2))
    assert sorted(topk.run()) == [("a", 4), ("i", 7)]


def test_stream_blocks():
    import heapq

    def map_topk(it):
        heap = []
        for symbol, count in it:
            heapq.heappush(heap, (count, symbol))
            if len(heap) > 2:
                heapq.heappop(heap)
        return ((1, x) for x in heap)

    def reduce_topk(it):
        counts = (v for _k, vit in it for v in vit)
        for count, symbol in heapq.nlargest(2, counts):
            yield symbol, count

    word = Dampr.memory(["supercalifragilisticexpialidociousa"])
    topk = word.flat_map(list).count() \
               .partition_map(map_topk) \
               .partition_reduce(reduce_topk)
    assert sorted(topk.run()) == [("a", 4), ("i", 7)]


def test_cross_map(items):
    item_counts = items.count()
    total = items.a_group_by(lambda x: 1, lambda x: 1).sum() \
                 .map(lambda x: float(x[1]))
    results = item_counts.cross_right(
        total, lambda ic, t: (ic[0], ic[1] / t)).read()
    assert sorted(results) == [(i, 1 / 10.0) for i in range(10, 20)]


def test_len(items):
    assert items.len().read() == [10]
    assert Dampr.memory([]).len().read() == [0]


def test_read_input():
    results = Dampr.read_input(RangeDataset(5), RangeDataset(10)) \
        .fold_by(lambda x: 1, lambda x, y: x + y).read()
    assert results[0][1] == sum(range(5)) + sum(range(10))


def test_file_glob():
    files = []
    for i in range(10):
        path = os.path.join("/tmp", "_test_dampr_amd_{}".format(i))
        with open(path, "w") as out:
            out.write(str(i))
        files.append(path)
    results = Dampr.text("/tmp/_test_dampr_amd_[135]") \
        .map(int).fold_by(lambda x: 1, lambda x, y: x + y).read()
    assert results == [(1, 1 + 3 + 5)]
    for fname in files:
        os.unlink(fname)


def test_topk():
    word = Dampr.memory(["supercalifragilisticexpialidociousa"])
    topk = word.flat_map(list).count().topk(5, lambda x: x[1])
    assert sorted(topk.run()) == [("a", 4), ("c", 3), ("i", 7),
                                  ("l", 3), ("s", 3)]


def test_file_links():
    dirnames = []
    for i in range(10):
        dirname = os.path.join("/tmp", "_test_dampr_amd_dir_{}".format(i))
        shutil.rmtree(dirname, ignore_errors=True)
        os.makedirs(dirname)
        dirnames.append(dirname)
        with open(os.path.join(dirname, "foo"), "w") as out:
            out.write(str(i))

    base = "/tmp/_dampr_amd_test_link"
    shutil.rmtree(base, ignore_errors=True)
    dirnames.append(base)
    os.makedirs(base)
    for i in (1, 3, 5):
        os.symlink(dirnames[i],
                   os.path.join(base, os.path.basename(dirnames[i])))

    results = Dampr.text(base).map(int) \
        .fold_by(lambda x: 1, lambda x, y: x + y).read()
    assert results == []

    results = Dampr.text(base, followlinks=True).map(int) \
        .fold_by(lambda x: 1, lambda x, y: x + y).read()
    assert results == [(1, 1 + 3 + 5)]

    for d in dirnames:
        shutil.rmtree(d)


def test_concat():
    word1 = Dampr.memory(list("abcdefg"))
    both = word1.concat(Dampr.memory(list("hijklmn")))
    assert sorted(both.run()) == list("abcdefghijklmn")


def test_map_values(items):
    results = items.map(lambda x: (x, x)) \
                   .map_values(lambda v: v + 1).read()
    assert results == [(i, i + 1) for i in range(10, 20)]


def test_map_keys(items):
    results = items.map(lambda x: (x, x)) \
                   .map_keys(lambda k: -k).read()
    assert results == [(-i, i) for i in range(10, 20)]


def test_prefix_suffix(items):
    assert items.prefix(lambda x: x % 2).read() == \
        [(i % 2, i) for i in range(10, 20)]
    assert items.suffix(lambda x: x % 2).read() == \
        [(i, i % 2) for i in range(10, 20)]


def test_mean():
    ages = [("Andrew", 33), ("Alice", 42), ("Andrew", 12), ("Bob", 51)]
    res = Dampr.memory(ages).mean(lambda x: x[0], lambda v: v[1]).read()
    assert res == [("Alice", 42.0), ("Andrew", 22.5), ("Bob", 51.0)]


def test_unique():
    names = [("Andrew", 1), ("Andrew", 1), ("Andrew", 2), ("Becky", 13)]
    res = Dampr.memory(names) \
        .group_by(lambda x: x[0], lambda x: x[1]).unique().read()
    assert res == [("Andrew", [1, 2]), ("Becky", [13])]


def test_sample_partial(items):
    got = items.sample(1.0).read()
    assert got == list(range(10, 20))


def test_filter_by_count():
    data = Dampr.memory([1, 1, 2, 2, 2, 3])
    res = filter_by_count(data, lambda x: x, lambda c: c >= 2).read()
    assert sorted(res) == [1, 1, 2, 2, 2]


def test_json_input(tmp_path):
    import json
    p = tmp_path / "data.json"
    with open(p, "w") as fh:
        for i in range(5):
            fh.write(json.dumps({"v": i}) + "\n")
    res = Dampr.json(str(p)).map(lambda d: d["v"]) \
        .fold_by(lambda x: 1, lambda x, y: x + y).read()
    assert res == [(1, 10)]


def test_reduce_buffer_spill(items):
    # reduce_buffer=1 forces a combine-dict flush per new key; results are
    # identical (the reference accepts but ignores this kwarg).
    out = items.a_group_by(lambda x: x % 2) \
               .reduce(lambda x, y: x + y, reduce_buffer=1).run()
    assert [kv[1] for kv in out] == [10 + 12 + 14 + 16 + 18,
                                     11 + 13 + 15 + 17 + 19]


def test_worker_crash_detected(items):
    # The reference deadlocks forever on a crashed worker
    # (stagerunner.py:35-37); we must raise instead.
    from dampr_amd.executor import WorkerCrash

    def boom(x):
        os._exit(13)

    with pytest.raises(Exception):
        items.map(boom).run()


def test_worker_exception_propagates(items):
    from dampr_amd.executor import WorkerCrash

    def raiser(x):
        raise ValueError("intentional")

    with pytest.raises(WorkerCrash, match="intentional"):
        items.map(raiser).run()


def test_uncompressed_spill(items):
    old = settings.compress_level
    settings.compress_level = 0
    try:
        assert list(items.map(lambda x: x * 2).run()) == \
            [i * 2 for i in range(10, 20)]
    finally:
        settings.compress_level = old


def test_inspect_passthrough(items, capsys):
    res = items.inspect("dbg").read()
    assert res == list(range(10, 20))


def test_indexer(tmp_path):
    from dampr_amd.utils.indexer import Indexer
    f = tmp_path / "data.txt"
    f.write_text("apple red\nbanana yellow\ncherry red\nkiwi green\n")
    idx = Indexer(str(f))
    idx.build(lambda line: line.split())
    red = sorted(idx.union(["red"]).read())
    assert red == ["apple red\n", "cherry red\n"]
    both = idx.intersect(["red", "apple"]).read()
    assert both == ["apple red\n"]
    either = sorted(idx.intersect(["red", "banana"], min_match=1).read())
    assert either == ["apple red\n", "banana yellow\n", "cherry red\n"]


def test_compaction_fan_in():
    """Force the file-count compaction pass: more map workers than
    max_files_per_stage means every partition exceeds the fan-in cap
    (reference: runner.py:293-320)."""
    from dampr_amd import settings as st
    old = st.max_files_per_stage
    st.max_files_per_stage = 3
    try:
        res = Dampr.memory(list(range(200)), partitions=16) \
            .count(lambda x: x % 5).run(n_maps=8)
        got = dict(res.read())
        assert got == {k: 40 for k in range(5)}
    finally:
        st.max_files_per_stage = old


def test_worker_exception_fails_fast():
    """A raising UDF must surface as WorkerCrash with the traceback, not
    hang the run (the reference deadlocks: stagerunner.py:35-37)."""
    from dampr_amd.executor import WorkerCrash

    def boom(x):
        raise ValueError("intentional-test-boom")

    with pytest.raises(WorkerCrash) as ei:
        Dampr.memory(list(range(100))).map(boom).count().run()
    assert "intentional-test-boom" in str(ei.value)


def test_worker_hard_death_detected():
    """A worker dying without reporting (os._exit) is detected instead of
    deadlocking."""
    import os as _os
    from dampr_amd.executor import WorkerCrash

    def die(x):
        _os._exit(13)

    with pytest.raises(WorkerCrash):
        Dampr.memory(list(range(100))).map(die).count().run()


def test_json_input(tmp_path):
    import json as json_mod
    f = tmp_path / "data.json"
    rows = [{"a": 1}, {"a": 2}, {"a": 5}]
    f.write_text("\n".join(json_mod.dumps(r) for r in rows) + "\n")
    got = sorted(Dampr.json(str(f)).map(lambda d: d["a"]).run().read())
    assert got == [1, 2, 5]


def test_gzip_text_input(tmp_path):
    import gzip
    f = tmp_path / "data.txt.gz"
    with gzip.open(str(f), "wt") as fh:
        fh.write("x\ny\nx\n")
    got = sorted(Dampr.text(str(f)).count().run().read())
    assert got == [("x", 2), ("y", 1)]


def test_urls_input_file_scheme(tmp_path):
    """UrlsInput through file:// URLs — the reference's live-network test
    (test_dampr.py:369-378) made offline-safe."""
    from dampr_amd.inputs import UrlsInput
    f = tmp_path / "remote.txt"
    f.write_text("hello\nworld\nhello\n")
    got = sorted(Dampr.read_input(
        UrlsInput(["file://" + str(f)])).count(lambda x: x.strip())
        .run().read())
    assert got == [("hello", 2), ("world", 1)]


def test_no_spill_file_leak(tmp_path):
    """After run + delete, no spill files remain (the reference leaks
    its pre-shuffle combine runs)."""
    res = Dampr.memory(list(range(1000))).count(lambda x: x % 7) \
        .run("leakcheck", working_dir=str(tmp_path))
    assert len(res.read()) == 7
    res.delete()
    leftover = []
    for root, _dirs, files in os.walk(str(tmp_path)):
        leftover.extend(os.path.join(root, f) for f in files)
    assert leftover == []


def test_reducer_exception_fails_fast():
    from dampr_amd.executor import WorkerCrash

    def bad_reduce(_k, vs):
        raise RuntimeError("reduce-boom")

    with pytest.raises(WorkerCrash) as ei:
        Dampr.memory(list(range(50))).group_by(lambda x: x % 3) \
            .reduce(bad_reduce).run()
    assert "reduce-boom" in str(ei.value)


def test_delete_prunes_empty_run_dirs(tmp_path):
    """ValueEmitter.delete removes the run's now-empty directory tree,
    not just the data files (ROADMAP 8: host-tier niceties)."""
    res = Dampr.memory(list(range(200))).count(lambda x: x % 5) \
        .run("prunecheck", working_dir=str(tmp_path))
    assert len(res.read()) == 5
    res.delete()
    assert not os.path.exists(os.path.join(str(tmp_path), "prunecheck"))


def test_multi_output_delete_keeps_sibling(tmp_path):
    """Deleting one of two retained outputs must not disturb the other."""
    words = Dampr.memory(["a b", "b c b"]).flat_map(lambda s: s.split())
    counts = words.count()
    total = counts.fold_by(lambda _wc: 1, lambda x, y: x + y,
                           value=lambda wc: wc[1])
    rc, rt = Dampr.run(counts, total, name="sibcheck",
                       working_dir=str(tmp_path))
    rc.delete()
    assert sorted(rt.read()) == [(1, 5)]
    rt.delete()
    assert not os.path.exists(os.path.join(str(tmp_path), "sibcheck"))


def test_mixed_type_join_never_hangs():
    """Joining int keys against string keys: cross-type keys are never
    equal, so the join is empty -- unless both types collide in one
    partition, where the comparison raises and the engine FAILS FAST
    (the reference deadlocks forever on the same pipeline: its crashed
    worker never posts to the result queue)."""
    from dampr_amd.executor import WorkerCrash
    try:
        got = Dampr.memory([1, 2, 3]).group_by(lambda v: v).join(
            Dampr.memory([("a", 1), ("b", 2)])
            .group_by(lambda kv: kv[0])) \
            .reduce(lambda l, r: (list(l), list(r))).run().read()
        assert got == []
    except WorkerCrash:
        pass                       # collision partition: fail-fast
