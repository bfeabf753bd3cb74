"""Module-global tunables.

API parity with the reference's ``dampr/settings.py`` (reference: settings.py:1-37):
users override by assignment, e.g. ``import dampr_amd.settings as settings;
settings.partitions = 128``.  GPU knobs are new (no analog in the reference,
which is CPU-only).
"""
import multiprocessing
import os

# ---------------------------------------------------------------- CPU engine

# Number of worker processes per stage pool (reference: settings.py:5).
max_processes = multiprocessing.cpu_count()

# zlib compression level for spilled runs.  0 (default) disables
# compression — measured ~20% faster end-to-end on NVMe-backed /tmp; the
# reference always gzips at level 1 (settings.py:8).  Set 1+ when spill
# space is tighter than spill bandwidth.
compress_level = 0

# Number of reduce partitions (reference: settings.py:11).
partitions = 91

# Cap on file fan-in per stage; larger sets are compacted by merge passes
# (reference: settings.py:16).
max_files_per_stage = 50

# Records per serialized frame in a spill file (reference: settings.py:20).
batch_size = 4000

# High-water RSS per worker process, in MB.  Crossing it flushes spill
# buffers to disk (reference: settings.py:27).
max_memory_per_worker = 512

# Adaptive memory-check pacing (reference: settings.py:31-37).  The governor
# estimates bytes/record and schedules the next RSS check; these bound it.
memory_min_count = 10000
memory_max_count_before_check = 100000

# Default cap on the map-side combine dictionary (distinct keys held
# before a forced spill).  A backstop under the RSS watermark: without it
# a high-cardinality a_group_by grows an unbounded per-worker dict
# between (amortized) RSS checks.  ARReduce.reduce's ``reduce_buffer``
# kwarg overrides per-stage (the reference documents that kwarg but never
# reads it — SURVEY.md §2.5).
reduce_buffer = 1 << 21

# ---------------------------------------------------------------- GPU engine
# New knobs for the MI355X path; no reference analog.

# Device-side record-batch size (records per columnar batch).
gpu_batch_records = 1 << 24

# Fraction of HBM the buffer pool may occupy before spilling to pinned host.
hbm_watermark = 0.90

# Hash-table load factor for the device combine table (K6).
gpu_table_load = 0.50

# Partitions per GPU for the device-side shuffle.
gpu_partitions_per_rank = 8

# Total partitions for the columnar engine's store (GpuRunner); sized so
# one partition of a 288 GB-per-GPU job fits comfortably in the pool.
gpu_partitions = 64

# Skewed-join guard: probe-side rows per hash-join batch.  A join
# partition larger than this is probed in chunks against the built
# table (inner/left; ROADMAP 7) so one hot key cannot blow out HBM.
gpu_join_probe_rows = 1 << 26

# Columnar engine HBM pool capacity (MB) before runs spill to pinned host.
# Default stays small enough for CPU test runs; bench/production set it to
# ~0.9 * free HBM.
hbm_pool_mb = int(os.environ.get("DAMPR_HBM_POOL_MB", "16384"))

# Directory for host-side spill of device batches.
spill_dir = os.environ.get("DAMPR_SPILL_DIR", "/tmp")

# Pinned-host spill tier capacity (MB); beyond it runs spill to raw files
# under ``spill_dir`` (NVMe tier).
host_pool_mb = int(os.environ.get("DAMPR_HOST_POOL_MB", "65536"))
