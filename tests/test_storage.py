"""Unit tests for the storage/spill tier and key hashing — coverage the
reference lacks entirely (it only has end-to-end tests)."""
import os

import pytest

from dampr_amd import settings
from dampr_amd.base import Splitter, FileSystem
from dampr_amd.dataset import (CombineSpiller, MergeDataset,
                               PartitionedSpiller, RunWriter, Spiller,
                               TextLineDataset)
from dampr_amd.keyhash import fnv1a64, key_hash64, partition_of, splitmix64


class _ZeroGov:
    """Governor that spills on every record."""

    def over_watermark(self):
        return True

    def reset(self):
        pass


@pytest.fixture
def fs(tmp_path):
    return FileSystem(str(tmp_path)).get_stage(0).get_worker(0)


def test_run_roundtrip(tmp_path):
    w = RunWriter(str(tmp_path / "run0"))
    kvs = [(i, "v{}".format(i)) for i in range(10000)]
    w.write_records(kvs)
    ds = w.close()
    assert list(ds.read()) == kvs
    ds.delete()
    assert not os.path.exists(ds.path)


def test_spiller_forced_spill(fs):
    sp = Spiller(fs, governor=_ZeroGov())
    data = [(k, k * 2) for k in (5, 3, 9, 1)]
    for k, v in data:
        sp.add_record(k, v)
    runs = sp.finished()[0]
    assert len(runs) == 4              # one run per record: forced spills
    merged = list(MergeDataset(runs).read())
    assert merged == sorted(data)


def test_spiller_sorts_runs(fs):
    sp = Spiller(fs)
    for k in [9, 2, 7, 1, 1, 4]:
        sp.add_record(k, k)
    runs = sp.finished()[0]
    assert len(runs) == 1
    assert [k for k, _ in runs[0].read()] == [1, 1, 2, 4, 7, 9]


def test_partitioned_spiller_sorted_within_partition(fs):
    sp = PartitionedSpiller(fs, Splitter(), 7)
    import random
    rng = random.Random(0)
    keys = [rng.randrange(1000) for _ in range(5000)]
    for k in keys:
        sp.add_record(k, k)
    parts = sp.finished()
    seen = []
    for p, runs in parts.items():
        for run in runs:
            ks = [k for k, _ in run.read()]
            assert ks == sorted(ks)
            assert all(hash(k) % 7 == p for k in ks)
            seen.extend(ks)
    assert sorted(seen) == sorted(keys)


def test_combine_spiller(fs):
    sp = CombineSpiller(fs, lambda a, b: a + b)
    for k in [1, 2, 1, 1, 2, 3]:
        sp.add_record(k, 1)
    runs = sp.finished()[0]
    assert list(MergeDataset(runs).read()) == [(1, 3), (2, 2), (3, 1)]


def test_combine_spiller_max_keys(fs):
    sp = CombineSpiller(fs, lambda a, b: a + b, max_keys=2)
    for k in [1, 2, 3, 1, 2, 3]:
        sp.add_record(k, 1)
    runs = sp.finished()[0]
    assert len(runs) > 1               # reduce_buffer cap forced spills
    total = {}
    for k, v in MergeDataset(runs).read():
        total[k] = total.get(k, 0) + v
    assert total == {1: 2, 2: 2, 3: 2}


def test_text_line_chunking(tmp_path):
    p = str(tmp_path / "t.txt")
    lines = ["line-{:04d}".format(i) for i in range(500)]
    with open(p, "w") as fh:
        fh.write("\n".join(lines) + "\n")
    size = os.path.getsize(p)
    # Chunk at an arbitrary boundary: every line exactly once.
    got = []
    step = 97
    for off in range(0, size, step):
        got.extend(v for _k, v in TextLineDataset(p, off, off + step).read())
    assert got == lines


def test_text_line_utf8(tmp_path):
    p = str(tmp_path / "u.txt")
    lines = ["héllo wörld {}".format(i) for i in range(50)]
    with open(p, "w", encoding="utf-8") as fh:
        fh.write("\n".join(lines) + "\n")
    size = os.path.getsize(p)
    got = []
    for off in range(0, size, 31):
        got.extend(v for _k, v in TextLineDataset(p, off, off + 31).read())
    assert got == lines


def test_key_hash_stability():
    # Pinned values: the HIP kernels implement the same functions and the
    # GPU numerics tests compare against these (ops/hip/common.h).
    assert fnv1a64(b"") == 0xcbf29ce484222325
    assert fnv1a64(b"a") == 0xaf63dc4c8601ec8c
    assert fnv1a64(b"hello") == 0xa430d84680aabd0b
    assert splitmix64(0) == 0xe220a8397b1dcdaf
    assert splitmix64(1) == 0x910a2dec89025cc1
    assert key_hash64("hello") == fnv1a64(b"hello")
    assert key_hash64(5) == splitmix64(5)
    assert 0 <= partition_of("anything", 91) < 91


def test_memory_governor_pacing():
    from dampr_amd.memory import MemoryGovernor
    gov = MemoryGovernor(limit_mb=1 << 30)   # never crossed
    spills = sum(gov.over_watermark() for _ in range(200000))
    assert spills == 0
