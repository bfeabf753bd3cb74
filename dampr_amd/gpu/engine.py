"""The device dataflow engine: executes DSL plans (runner.Graph) over
columnar record batches resident in HBM.

Architecture (MI355X-first, no translation of the reference's process
pools):

* Records are **typed columns** — an i64 key column + an i64/f64 value
  column — not streams of pickled Python objects.  This is the layout the
  CDNA4 kernels (ops/hip/*.hip) operate on: radix sort (K2/K3), segmented
  reduce (K5/K7), hash join (K8), top-k (K11).  Pipelines over arbitrary
  Python objects belong to the host engine (runner.MTRunner), which is the
  complete, conformance-tested implementation of the same plans; pipelines
  over text use the tokenizer/string-dict specialization (gpu/tfidf.py).
* A stage lowers to device kernels when the DSL built it from *recognized*
  functions (dampr_amd.funcs); anything else falls back per-stage to the
  host operators with decode/encode at the boundary (SURVEY.md §7 "hard
  parts": opaque UDFs run on the host, the shuffle/sort/combine core stays
  on device).
* Out-of-core: runs live in an ``HbmPool`` with two watermarks; least
  recently used runs spill to pinned host memory and on to raw NVMe
  files, paging back on touch — the reference's RSS-watermark spill
  files (dampr/memory.py, dataset.py:190-262) re-expressed over HBM →
  host → disk tiers.  Single-rank columnar ingest is *lazy*: batched
  unrouted runs, streamed run-by-run through record-wise stages; only
  key-colocating stages (reduce/join) route to hash partitions, and a
  skewed partition larger than half the pool reduces run-by-run with an
  associative re-reduce.  Only the active working set must be resident,
  so jobs scale past device memory.
* Multi-GPU: after every partitioning map stage the engine exchanges
  partitions to their owning rank over RCCL all-to-all (xGMI); partition p
  is owned by rank ``p % world`` (parallel/shuffle.py).
"""
import logging
import os

import torch

from .. import settings
from ..runner import GMap, GReduce, GSink, RunnerBase
from .backend import ops_for

log = logging.getLogger("dampr_amd")


# --------------------------------------------------------------------------
# Columnar data
# --------------------------------------------------------------------------

_VAL_DTYPES = (torch.int64, torch.float64)


def _as_column(x, device=None, dtype=None):
    """device=None keeps a torch input where it lives (an HBM-resident
    tensor must not bounce through the host)."""
    if isinstance(x, torch.Tensor):
        t = x
    else:
        import numpy as np
        t = torch.from_numpy(np.ascontiguousarray(x))
    if t.dtype not in _VAL_DTYPES:
        t = t.to(torch.float64 if t.is_floating_point() else torch.int64)
    if dtype is not None:
        t = t.to(dtype)
    return t if device is None else t.to(device)


class ColumnSource(object):
    """A device-columnar input: (keys i64, vals i64|f64).  String key
    arrays dictionary-encode at ingest (np.unique: sorted uniques +
    inverse ids, so rank order == lexicographic order); the table is
    built from the FULL input before any per-rank slicing, so every
    rank shares one dictionary."""

    dampr_columnar = True          # engine-selection sentinel (dampr.py)

    def __init__(self, keys, vals, str_table=None, fkeys=False):
        assert keys.dtype == torch.int64
        assert getattr(vals, "is_strvals", False) \
            or vals.dtype in _VAL_DTYPES
        assert keys.numel() == vals.numel()
        self.keys = keys
        self.vals = vals
        self.str_table = str_table
        # keys hold the f64 order-preserving encode (decoded on read)
        self.fkeys = fkeys

    @staticmethod
    def _is_str_array(x):
        import numpy as np
        if isinstance(x, np.ndarray):
            return x.dtype.kind in ("U", "S") or (
                x.dtype.kind == "O" and x.size
                and all(isinstance(e, str) for e in x.flat))
        return (isinstance(x, (list, tuple)) and len(x)
                and all(isinstance(e, str) for e in x))

    @classmethod
    def from_data(cls, vals, keys=None, device=None):
        import numpy as np
        if cls._is_str_array(vals):
            # var-len value arena: strings move as opaque bytes
            # addressed by row; only keys compare on device
            from .strvals import StrVals
            arr = np.asarray(vals)
            if arr.dtype.kind == "S":
                arr = arr.astype("U")
            v = StrVals.from_strings([str(s) for s in arr.flat],
                                     device=device)
        else:
            v = _as_column(vals, device)
        str_table = None
        fkeys = False
        if keys is None:
            k = torch.arange(v.numel(), dtype=torch.int64, device=v.device)
        elif (isinstance(keys, torch.Tensor)
              and keys.dtype.is_floating_point) or (
                  isinstance(keys, np.ndarray)
                  and keys.dtype.kind == "f"):
            # float keys: order-preserving f64 encode, NOT an int cast
            # (which would silently truncate 1.5 and 2.5 into key 1)
            kf = keys.to(torch.float64) if isinstance(keys, torch.Tensor) \
                else torch.from_numpy(keys.astype(np.float64))
            if kf.device != v.device:
                kf = kf.to(v.device)
            k = _encode_f64_sortable(kf)
            fkeys = True
        elif cls._is_str_array(keys):
            arr = np.asarray(keys)
            if arr.dtype.kind == "S":
                arr = arr.astype("U")
            uniq, inv = np.unique(arr, return_inverse=True)
            str_table = tuple(str(u) for u in uniq)
            k = torch.from_numpy(inv.astype(np.int64))
            if device is not None or k.device != v.device:
                k = k.to(v.device)
        else:
            k = _as_column(keys, device, torch.int64)
            if k.device != v.device:
                k = k.to(v.device)
        return cls(k, v, str_table=str_table, fkeys=fkeys)


class TextSource(object):
    """Newline-delimited text input for the device engine: the
    ``device_text(...).flat_map(funcs.tokenize_set).count()`` idiom lowers
    onto the fused single-pass document-frequency kernel (gpu/tfidf.py)."""

    dampr_columnar = True

    def __init__(self, data):
        import numpy as np
        self.text_t = None             # device-resident u8 tensor, if given
        self._text_np = None
        if isinstance(data, torch.Tensor):
            assert data.dtype == torch.uint8
            self.text_t = data
        elif isinstance(data, str):
            with open(data, "rb") as fh:
                raw = fh.read()
            self._text_np = np.frombuffer(raw, dtype=np.uint8).copy()
        elif isinstance(data, bytes):
            self._text_np = np.frombuffer(data, dtype=np.uint8).copy()
        else:
            self._text_np = np.asarray(data, dtype=np.uint8)

    @property
    def nbytes(self):
        return (self.text_t.numel() if self.text_t is not None
                else self._text_np.nbytes)

    @property
    def text(self):
        """Host view (materialized on demand for fallback paths)."""
        if self._text_np is None:
            self._text_np = self.text_t.cpu().numpy()
        return self._text_np


class TokenStore(object):
    """Result of the fused text document-frequency stage: token-hash keys
    + counts, with the string dictionary needed to materialize tokens."""

    keyed = False
    fkeys = False

    def __init__(self, engine, keys, vals, text_dev):
        self.engine = engine
        self.keys = keys
        self.vals = vals
        self.text_dev = text_dev


class TokenColumnDataset(object):
    """ColumnDataset analog whose keys decode to token strings via the
    device string dictionary."""

    def __init__(self, store, keyed):
        self.store = store
        self.keyed = keyed

    def read(self):
        st = self.store
        blob, lens = st.engine.token_strings(st.keys, st.text_dev)
        vals = st.vals.cpu().tolist()
        out = []
        pos = 0
        b = blob.tobytes()
        for i, ln in enumerate(lens):
            tok = b[pos:pos + int(ln)].decode("ascii")
            pos += int(ln)
            if self.keyed:
                out.append((tok, (tok, vals[i])))
            else:
                out.append((tok, vals[i]))
        out.sort(key=lambda r: r[0])
        return iter(out)

    def grouped_read(self):
        import itertools
        for key, group in itertools.groupby(self.read(),
                                            key=lambda p: p[0]):
            yield key, (v for _k, v in group)

    def delete(self):
        self.store = None

    def __iter__(self):
        return self.read()


class PartStore(dict):
    """{partition -> [DeviceRun]} plus column metadata:

    * ``keyed``: output of a keyed reducer — records follow the host
      convention (k, (k, v)) at decode time (base.py KeyedReduce).
    * ``fkeys``: the key column is an order-preserving i64 encoding of
      float64 keys (relational.encode_f64_sortable); decoded on read.
    """

    def __init__(self, keyed=False, fkeys=False, partitioned=True,
                 str_table=None, svals=False):
        super(PartStore, self).__init__()
        self.keyed = keyed
        self.fkeys = fkeys
        # value columns are var-len byte arenas (StrVals).  Kept as a
        # STORE-level flag (not just per-run inspection) so an empty
        # rank's store still reports it: stage-dispatch guards must
        # agree across ranks or collective sequences desync.
        self.svals = svals
        # dictionary-encoded string keys: key column holds ranks into
        # this sorted tuple (rank order == lexicographic order, so sorted
        # device output decodes to host-ordered strings).  Cross-store
        # combinations remap ids into the union table on device
        # (_unify_str_stores); only dictionary + dictionary-less mixes
        # fall back to host records (_decode_store).
        self.str_table = str_table
        # False = runs in pseudo-partition 0, not yet routed by key hash
        # (lazy ingest: stages that need co-partitioned data call
        # _ensure_partitioned; record-wise maps stream run by run)
        self.partitioned = partitioned


class HostStore(list):
    """Fallback record storage: a plain list of (k, v) Python records for
    stages whose outputs don't fit typed columns.  Downstream stages run
    on the host until records become numeric again (then they re-enter the
    columnar path via _encode_or_host)."""

    keyed = False
    fkeys = False


class SinkStore(object):
    """A sink stage's output: lazy view over the durable text part
    files it wrote (host parity — _SinkWorker returns its
    TextLineDatasets, so a sink's output is readable downstream without
    re-materializing the sunk bytes; reference: the sink's
    TextLineDataset handles flow back as normal data,
    dampr/dataset.py:280-282, runner.py:195-197)."""

    keyed = False
    fkeys = False

    def __init__(self, paths):
        self.paths = paths

    def datasets(self):
        from ..dataset import TextLineDataset
        return [TextLineDataset(p) for p in self.paths]


def _is_sv(x):
    """Is this value column a var-len byte arena (gpu.strvals.StrVals)?"""
    return getattr(x, "is_strvals", False)


def _cat_vals(vs):
    """Concatenate value columns (tensor or StrVals, never mixed)."""
    if len(vs) == 1:
        return vs[0]
    if any(_is_sv(v) for v in vs):
        from .strvals import StrVals
        assert all(_is_sv(v) for v in vs), \
            "mixed var-len and numeric value runs in one partition"
        return StrVals.cat(vs)
    return torch.cat(vs)


def _run_has_sv(run):
    """Does this run carry var-len values (any tier)?  Reads snapshot
    attributes (IO threads may migrate the run between tiers)."""
    vals = run.vals
    if run.keys is not None and vals is not None:
        return _is_sv(vals)
    host = run._host
    if host is not None:
        return _is_sv(host[1])
    meta = run._meta
    return meta is not None and isinstance(meta[1], tuple)


def _store_has_sv(store):
    if not isinstance(store, PartStore):
        return False
    return getattr(store, "svals", False) or any(
        _run_has_sv(r) for runs in store.values() for r in runs)


def _decode_f64_sortable(enc):
    """Inverse of relational.encode_f64_sortable."""
    sign_bit = -(1 << 63)
    b = torch.where(enc < 0, enc ^ sign_bit, ~enc)
    return b.view(torch.float64)


def _encode_f64_sortable(x):
    # + 0.0 canonicalizes -0.0 to +0.0 (Python == merges them as one
    # group key; distinct bit patterns would split it)
    b = (x + 0.0).view(torch.int64)
    sign_bit = -(1 << 63)
    return torch.where(b < 0, ~b, b ^ sign_bit)


class _PinnedPool(object):
    """Recycled pinned-host buffers for the spill tier: cudaHostAlloc of
    a multi-hundred-MB buffer costs tens of ms, and a >pool job spills
    hundreds of runs.  Power-of-two u8 buckets, views carved to size;
    a buffer with an in-flight H2D reload is only reused after its
    event completes."""

    def __init__(self, cap_bytes=192 << 30):
        import threading
        self.free = {}                 # bucket bytes -> [u8 base]
        self.free_bytes = 0
        self.cap = cap_bytes
        self._pending = []             # (base, bucket, event)
        self._lock = threading.Lock()  # IO threads stage into the pool

    @staticmethod
    def _bucket(nbytes):
        return 1 << max(12, int(nbytes - 1).bit_length())

    def _drain(self):
        still = []
        for base, b, evt in self._pending:
            if evt is not None and not evt.query():
                still.append((base, b, evt))
                continue
            if self.free_bytes + b <= self.cap:
                self.free.setdefault(b, []).append(base)
                self.free_bytes += b
        self._pending = still

    def get_like(self, t):
        """Pinned host tensor with t's shape/dtype."""
        return self.get(t.numel(), t.dtype)

    def get(self, numel, dtype):
        nbytes = numel * torch._utils._element_size(dtype)
        pin = torch.cuda.is_available()
        if nbytes == 0:
            return torch.empty(0, dtype=dtype, pin_memory=pin)
        with self._lock:
            self._drain()
            b = self._bucket(nbytes)
            lst = self.free.get(b)
            if lst:
                base = lst.pop()
                self.free_bytes -= b
            else:
                base = None
        if base is None:
            base = torch.empty(self._bucket(nbytes), dtype=torch.uint8,
                               pin_memory=pin)
        return base[:nbytes].view(dtype)

    def put(self, t, event=None):
        """Return a buffer; with ``event``, reuse waits for it (an
        async H2D may still be reading the pinned memory)."""
        if t is None or not isinstance(t, torch.Tensor) \
                or not t.is_pinned():
            return
        base = t
        while getattr(base, "_base", None) is not None:
            base = base._base
        nb = base.numel() * base.element_size()
        if base.dtype != torch.uint8 or nb != self._bucket(nb):
            return                     # not one of our bucket bases
        with self._lock:
            self._pending.append((base, nb, event))


_PIN = _PinnedPool()


class DeviceRun(object):
    """One (keys, vals) run of a partition, spillable down the tier
    hierarchy HBM -> pinned host -> NVMe file (the reference's gzip-pickle
    /tmp spill files, dataset.py:119-188, re-expressed for 288 GB HBM +
    host DRAM + NVMe)."""

    __slots__ = ("keys", "vals", "sorted", "_host", "_disk", "_meta",
                 "_evt", "_dfut", "_lfut")

    def __init__(self, keys, vals, sorted=False):
        self.keys = keys
        self.vals = vals
        self.sorted = sorted
        self._host = None
        self._disk = None
        self._meta = None
        self._evt = None           # in-flight async D2H spill marker
        self._dfut = None          # in-flight threaded disk WRITE
        self._lfut = None          # in-flight threaded disk READ-AHEAD

    @property
    def n(self):
        """Row count, available without paging the run in."""
        if self.keys is not None:
            return self.keys.numel()
        if self._meta is not None:
            return self._meta[0]
        return 0

    @property
    def nbytes(self):
        if self.keys is not None:
            if _is_sv(self.vals):
                return self.keys.numel() * 8 + self.vals.nbytes
            return self.keys.numel() * 8 + self.vals.element_size() * \
                self.vals.numel()
        if self._meta is not None:
            n, vdt = self._meta
            if isinstance(vdt, tuple):          # ("str", blob_bytes)
                return n * 8 + vdt[1] + (n + 1) * 8
            return n * 8 + n * (8 if vdt in (torch.int64, torch.float64)
                                else 8)
        return 0

    @property
    def resident(self):
        return self.keys is not None

    @property
    def on_disk(self):
        return self._disk is not None

    def _wait_spill(self):
        """Block until an in-flight async spill's D2H copies land."""
        if self._evt is not None:
            self._evt.synchronize()
            self._evt = None

    def _wait_io(self):
        """Join any threaded disk write/read touching this run."""
        if self._dfut is not None:
            self._dfut.result()
            self._dfut = None
        if self._lfut is not None:
            self._lfut.result()
            self._lfut = None

    def _stage_host(self):
        """IO-thread body: page the run's bytes from NVMe into pinned
        host buffers (read-ahead; the H2D happens at touch time).
        Joins a pending disk WRITE first (executor FIFO guarantees the
        write already started, so this cannot self-deadlock)."""
        self._wait_spill()
        if self._dfut is not None:
            self._dfut.result()
            self._dfut = None
        self._load_host()

    def drop(self):
        """Release storage on every tier (caller owns accounting and
        disk unlink).  Waits for in-flight spill DMA first — freeing a
        pinned buffer under an active copy corrupts host memory."""
        self._wait_spill()
        self._wait_io()
        if self._host is not None:
            hk, hv = self._host
            _PIN.put(hk)
            if _is_sv(hv):
                _PIN.put(hv.blob)
                _PIN.put(hv.offs)
            else:
                _PIN.put(hv)
        self.keys = None
        self.vals = None
        self._host = None
        self._meta = None

    def writeback_async(self, stream):
        """Copy HBM -> pinned host in the background WITHOUT evicting:
        runs are immutable, so the host copy never goes stale and a
        later eviction becomes a pointer drop instead of a synchronous
        (or burst-clustered) D2H on the critical path."""
        if stream is None or self.keys is None or self._host is not None \
                or self.keys.device.type != "cuda":
            return
        from .strvals import StrVals
        main = torch.cuda.current_stream(self.keys.device)
        with torch.cuda.stream(stream):
            # copies must see the producing kernels' writes
            stream.wait_stream(main)
            hk = _PIN.get_like(self.keys)
            hk.copy_(self.keys, non_blocking=True)
            self.keys.record_stream(stream)
            if _is_sv(self.vals):
                hb = _PIN.get_like(self.vals.blob)
                ho = _PIN.get_like(self.vals.offs)
                hb.copy_(self.vals.blob, non_blocking=True)
                ho.copy_(self.vals.offs, non_blocking=True)
                self.vals.record_stream(stream)
                hv = StrVals(hb, ho)
                self._meta = (self.keys.numel(),
                              ("str", self.vals.blob.numel()))
            else:
                hv = _PIN.get_like(self.vals)
                hv.copy_(self.vals, non_blocking=True)
                self.vals.record_stream(stream)
                self._meta = (self.keys.numel(), self.vals.dtype)
            evt = torch.cuda.Event()
            evt.record(stream)
        self._evt = evt
        self._host = (hk, hv)

    @property
    def clean(self):
        """Resident with a (possibly in-flight) host copy."""
        return self.keys is not None and self._host is not None

    def drop_device(self):
        """Evict a CLEAN run: the host copy exists, so releasing the
        device tensors is the whole eviction (record_stream at copy
        time defers allocator reuse until the D2H lands)."""
        assert self._host is not None
        self.keys = None
        self.vals = None

    def spill_async(self, stream):
        """HBM -> pinned host on a dedicated D2H stream, overlapped
        with compute on the main stream.  Device tensors are released
        immediately (record_stream defers allocator reuse until the
        copies complete); host-side readers must _wait_spill()."""
        if stream is None or self.keys is None \
                or self.keys.device.type != "cuda":
            return self.spill()
        if self._host is not None:          # clean: already copied
            self.keys = None
            self.vals = None
            return
        self.writeback_async(stream)
        self.keys = None
        self.vals = None

    def spill(self):
        """HBM -> (pinned) host memory."""
        if self._host is not None or self.keys is None:
            return
        pin = self.keys.device.type == "cuda"

        def _halloc(t):
            return _PIN.get_like(t) if pin else \
                torch.empty_like(t, device="cpu")

        hk = _halloc(self.keys)
        hk.copy_(self.keys)
        if _is_sv(self.vals):
            from .strvals import StrVals
            hb = _halloc(self.vals.blob)
            ho = _halloc(self.vals.offs)
            hb.copy_(self.vals.blob)
            ho.copy_(self.vals.offs)
            hv = StrVals(hb, ho)
            self._meta = (self.keys.numel(),
                          ("str", self.vals.blob.numel()))
        else:
            hv = _halloc(self.vals)
            hv.copy_(self.vals)
            self._meta = (self.keys.numel(), self.vals.dtype)
        self._host = (hk, hv)
        self.keys = None
        self.vals = None

    def spill_to_disk(self, path):
        """Host -> NVMe file (raw little-endian columns, no pickle)."""
        if self._host is None:
            return
        self._wait_spill()
        hk, hv = self._host
        with open(path, "wb") as fh:
            hk.numpy().tofile(fh)
            if _is_sv(hv):
                hv.offs.numpy().tofile(fh)
                hv.blob.numpy().tofile(fh)
            else:
                hv.numpy().tofile(fh)
        self._disk = path
        self._host = None
        _PIN.put(hk)
        if _is_sv(hv):
            _PIN.put(hv.blob)
            _PIN.put(hv.offs)
        else:
            _PIN.put(hv)

    def _load_host(self):
        if self._host is None and self._disk is not None:
            n, vdt = self._meta

            def _readinto(fh, numel, dtype):
                t = _PIN.get(numel, dtype)
                if numel:
                    mv = memoryview(t.numpy()).cast("B")
                    got = fh.readinto(mv)
                    assert got == len(mv), "short spill-file read"
                return t

            with open(self._disk, "rb") as fh:
                hk = _readinto(fh, n, torch.int64)
                if isinstance(vdt, tuple):
                    from .strvals import StrVals
                    ho = _readinto(fh, n + 1, torch.int64)
                    hb = _readinto(fh, vdt[1], torch.uint8)
                    hv = StrVals(hb, ho)
                else:
                    hv = _readinto(fh, n, vdt)
            os_mod = __import__("os")
            try:
                os_mod.unlink(self._disk)
            except OSError:
                pass
            self._disk = None
            self._host = (hk, hv)

    def load(self, device):
        if self.keys is None:
            self._wait_spill()
            self._wait_io()
            self._load_host()
            hk, hv = self._host
            self.keys = hk.to(device, non_blocking=True)
            self.vals = hv.to(device, non_blocking=True)
            self._host = None
            if torch.device(device).type == "cuda":
                evt = torch.cuda.Event()
                evt.record(torch.cuda.current_stream(device))
                _PIN.put(hk, evt)
                if _is_sv(hv):
                    _PIN.put(hv.blob, evt)
                    _PIN.put(hv.offs, evt)
                else:
                    _PIN.put(hv, evt)
        return self


class HbmPool(object):
    """Two-watermark tier governor over run bytes (the reference's
    MemoryChecker/MaxMemoryWriter analog, memory.py:72-113, re-expressed
    over tiers): device-resident bytes above ``capacity`` spill to pinned
    host; host-resident bytes above ``host_capacity`` spill on to raw
    NVMe files under ``spill_dir``."""

    def __init__(self, capacity_bytes, host_capacity=None, spill_dir=None):
        import os
        import uuid
        self.capacity = capacity_bytes
        self.used = 0
        # ordered sets (insertion-ordered dicts): O(1) add/remove/contains
        # — a 2 TB job at 128 MB runs is ~16k live runs, list.remove per
        # touch would be O(n) scans on every access
        self._lru = {}                 # device-resident, insertion order
        self.host_capacity = (host_capacity
                              if host_capacity is not None
                              else float("inf"))
        self.host_used = 0
        self._host_lru = {}
        self.spill_dir = spill_dir or settings.spill_dir
        # dedicated D2H stream: evictions overlap main-stream compute
        # (set by the engine on CUDA devices; None = synchronous spill)
        self.spill_stream = None
        self._run_tag = "dampr_amd_{}".format(uuid.uuid4().hex[:10])
        self._file_ctr = 0
        self._disk_paths = []
        self.spilled_host = 0
        self.spilled_disk = 0
        self.reloaded = 0
        self.clean_bytes = 0       # resident runs with a host copy
        from collections import deque
        self._wb_queue = deque()   # writeback candidates, admit order
        self._os = os
        self._io = None            # lazy ThreadPoolExecutor (NVMe IO)

    def _executor(self):
        if self._io is None:
            from concurrent.futures import ThreadPoolExecutor
            self._io = ThreadPoolExecutor(max_workers=2)
        return self._io

    def _next_path(self):
        self._os.makedirs(self.spill_dir, exist_ok=True)
        self._file_ctr += 1
        p = self._os.path.join(
            self.spill_dir,
            "{}_{}.run".format(self._run_tag, self._file_ctr))
        self._disk_paths.append(p)
        return p

    def cleanup(self):
        """Unlink any spill files still on disk (runs never paged back
        before the job finished)."""
        if self._io is not None:
            self._io.shutdown(wait=True)
            self._io = None
        for p in self._disk_paths:
            try:
                self._os.unlink(p)
            except OSError:
                pass
        self._disk_paths = []

    def stats(self):
        return {"hbm_used": self.used, "host_used": self.host_used,
                "spilled_to_host_bytes": self.spilled_host,
                "spilled_to_disk_bytes": self.spilled_disk,
                "reloads_bytes": self.reloaded}

    def admit(self, run):
        self.used += run.nbytes
        self._lru[run] = None
        self._wb_queue.append(run)
        self.balance()

    def prefetch(self, runs, device, stream):
        """Begin paging spilled runs back on a side HIP stream (overlaps
        the H2D copies with the current partition's compute).  The caller
        must make its stream wait on ``stream`` before consuming."""
        if stream is None:
            return
        main = torch.cuda.current_stream(device)
        # disk-resident runs: stage file -> pinned on the IO thread NOW
        # (the H2D happens at touch); host-resident runs take the H2D
        # side-stream path below
        for run in runs:
            if run.resident or run._lfut is not None:
                continue
            if run.on_disk or run._dfut is not None:
                run._lfut = self._executor().submit(run._stage_host)
        with torch.cuda.stream(stream):
            for run in runs:
                if run.resident or run._lfut is not None \
                        or run._dfut is not None or run.on_disk:
                    continue
                if run in self._host_lru:
                    del self._host_lru[run]
                    self.host_used -= run.nbytes
                self.reloaded += run.nbytes
                run.load(device)
                # tensors are allocated on the side stream but consumed
                # on the main stream: mark the cross-stream use so the
                # caching allocator does not reuse them early
                run.keys.record_stream(main)
                run.vals.record_stream(main)
                self.used += run.nbytes
                self._lru[run] = None
        # NOTE: no balance() here — eviction during an in-flight copy
        # could spill the very runs being loaded; the next touch() call
        # rebalances on the main stream.

    def touch(self, run, device):
        """Page the run in (if spilled) and refresh it to MRU.  ``used``
        counts every tracked resident run — round 1's model decremented
        on release() and double-decremented on repeated touch/release
        cycles, driving ``used`` negative and disabling eviction."""
        if not run.resident:
            if run in self._host_lru:
                del self._host_lru[run]
                self.host_used -= run.nbytes
            self.reloaded += run.nbytes
            run.load(device)
            self.used += run.nbytes
            self._lru[run] = None
            self._wb_queue.append(run)
            self.balance(exclude=run)
        elif run in self._lru:
            del self._lru[run]          # MRU refresh
            self._lru[run] = None
        return run

    def release(self, run):
        """The caller is done with the run for this stage.  Accounting
        is unchanged — the run stays tracked and evictable (an evicted
        run's tensors survive through the caller's own references until
        it drops them)."""

    def forget(self, run):
        """Remove a run from all tier tracking (it is being freed)."""
        if run in self._lru:
            del self._lru[run]
            if run.resident:
                self.used -= run.nbytes
                if run._host is not None:
                    self.clean_bytes -= run.nbytes
        if run in self._host_lru:
            del self._host_lru[run]
            self.host_used -= run.nbytes

    def balance(self, exclude=None):
        # background WRITEBACK above the half-full watermark: LRU runs
        # copy to pinned host while still resident (runs are immutable,
        # so the copy never goes stale); their later eviction is a
        # pointer drop.  Without this, a larger pool defers all D2H
        # into bursts that serialize against the reduce phase's
        # reloads (measured 2x wall at 120 GB with a 32 GB pool).
        if self.spill_stream is not None \
                and self.used > self.capacity // 2 \
                and self.clean_bytes * 2 < self.capacity:
            # FIFO candidate queue (admit order ~ LRU): O(1) amortized
            # — scanning the LRU dict would walk its clean prefix on
            # every admit
            wrote = 0
            while wrote < 4 and self._wb_queue \
                    and self.clean_bytes * 2 < self.capacity:
                r = self._wb_queue.popleft()
                if r is exclude:
                    self._wb_queue.append(r)
                    break
                if not r.resident or r._host is not None \
                        or r not in self._lru:
                    continue            # freed/evicted/already clean
                r.writeback_async(self.spill_stream)
                self.clean_bytes += r.nbytes
                wrote += 1
        while self.used > self.capacity and self._lru:
            victim = None
            for r in self._lru:
                if r is not exclude and r.resident:
                    victim = r
                    break
            if victim is None:
                return
            del self._lru[victim]
            self.used -= victim.nbytes
            if victim._host is not None:
                self.clean_bytes -= victim.nbytes
            victim.spill_async(self.spill_stream)
            self.spilled_host += victim.nbytes
            self.host_used += victim.nbytes
            self._host_lru[victim] = None
        while self.host_used > self.host_capacity and self._host_lru:
            v = next(iter(self._host_lru))
            del self._host_lru[v]
            self.host_used -= v.nbytes
            self.spilled_disk += v.nbytes
            path = self._next_path()
            if self.spill_stream is None:
                v.spill_to_disk(path)      # CPU/test path stays sync
            else:
                # NVMe writes ride an IO thread, overlapped with
                # compute; readers join via _wait_io
                v._dfut = self._executor().submit(v.spill_to_disk, path)


# --------------------------------------------------------------------------
# Output dataset (feeds ValueEmitter / downstream host code)
# --------------------------------------------------------------------------

class ColumnDataset(object):
    """Dataset-duck-typed view over result columns: read() yields (k, v)
    Python scalars; columns() hands back the tensors for zero-copy
    composition.  ``keyed`` reproduces the host reducers' value
    convention (k, (k, v)); ``fkeys`` decodes float64 keys."""

    def __init__(self, keys, vals, keyed=False, fkeys=False,
                 str_table=None):
        self.keys_t = keys
        self.vals_t = vals
        self.keyed = keyed
        self.fkeys = fkeys
        self.str_table = str_table

    def columns(self):
        return self.keys_t, self.vals_t

    def read(self):
        kt = self.keys_t
        if self.fkeys:
            kt = _decode_f64_sortable(kt)
        k = kt.cpu().tolist()
        if self.str_table is not None:
            tbl = self.str_table
            k = [tbl[i] for i in k]
        v = self.vals_t.cpu().tolist()
        if self.keyed:
            return iter((kk, (kk, vv)) for kk, vv in zip(k, v))
        return iter(zip(k, v))

    def grouped_read(self):
        import itertools
        for key, group in itertools.groupby(self.read(), key=lambda p: p[0]):
            yield key, (v for _k, v in group)

    def delete(self):
        self.keys_t = self.vals_t = None

    def __iter__(self):
        return self.read()


# --------------------------------------------------------------------------
# The engine
# --------------------------------------------------------------------------

class GpuRunner(RunnerBase):
    """Interprets a runner.Graph over device columns.

    Stage dispatch: ``stage.options["device_map"] / ["device_reduce"]``
    descriptors (attached by the DSL when built from recognized funcs) run
    on the kernels; untagged stages run the host operators with a
    decode/encode boundary (records must stay numeric scalars).
    """

    def __init__(self, name, graph, device=None, n_partitions=None,
                 hbm_bytes=None, host_bytes=None, spill_dir=None):
        # note: RunnerBase.__init__ builds a /tmp FileSystem we don't use;
        # keep it for interface parity (sinks reuse its naming).
        super(GpuRunner, self).__init__(name, graph)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.ops = ops_for(self.device)
        cap = hbm_bytes or settings.hbm_pool_mb * (1 << 20)
        host_cap = host_bytes if host_bytes is not None \
            else settings.host_pool_mb * (1 << 20)
        self.pool = HbmPool(cap, host_capacity=host_cap,
                            spill_dir=spill_dir)
        if torch.distributed.is_available() and \
                torch.distributed.is_initialized():
            self.world = torch.distributed.get_world_size()
            self.rank = torch.distributed.get_rank()
        else:
            self.world, self.rank = 1, 0
        self.exchanged_rows = 0
        self._side_stream = (torch.cuda.Stream(device=self.device)
                             if self.device.type == "cuda" else None)
        # separate D2H stream: spills and prefetches use the DMA
        # engines in both directions concurrently with compute
        self.pool.spill_stream = (torch.cuda.Stream(device=self.device)
                                  if self.device.type == "cuda" else None)
        if n_partitions:
            self.n_partitions = n_partitions
        elif self.world == 1 and self._inputs_fit(cap):
            # single rank, everything resident: one partition means one
            # full-device sort/reduce per stage instead of 64 small ones
            # (the kernels are whole-device parallel; partitions exist
            # for out-of-core paging and the cross-rank exchange)
            self.n_partitions = 1
        else:
            self.n_partitions = settings.gpu_partitions

    def _inputs_fit(self, cap):
        total = 0
        for inp in self.graph.inputs.values():
            if isinstance(inp, ColumnSource):
                total += inp.keys.numel() * 8 + \
                    inp.vals.element_size() * inp.vals.numel()
            elif isinstance(inp, TextSource):
                total += inp.nbytes
            else:
                return False             # unknown size: keep partitions
        return total * 4 < cap           # headroom for intermediates

    # -- plan walk ---------------------------------------------------------

    def run(self, outputs, cleanup=True):
        from ..utils.trace import get_trace, trace_stage
        get_trace().clear()
        data = {}
        # consumer refcounts: a store is freed after its LAST consuming
        # stage (keeps true >HBM jobs from pinning dead intermediates)
        consumers = {}
        for stage in self.graph.stages:
            for src in stage.inputs:
                consumers[src] = consumers.get(src, 0) + 1
        for src in outputs:
            consumers[src] = consumers.get(src, 0) + 1
        for src, inp in self.graph.inputs.items():
            with trace_stage("ingest {}".format(src), self.device):
                data[src] = self._ingest(inp)
        for stage_id, stage in enumerate(self.graph.stages):
            log.info("[device] Stage %s/%s: %r", stage_id + 1,
                     len(self.graph.stages), stage)
            ins = [data[i] for i in stage.inputs]
            # consume-as-you-go: inputs whose LAST consumer is this
            # stage may free each partition right after it is reduced /
            # joined (out-of-core jobs would otherwise re-evict dead
            # partition data until stage-end cleanup)
            self._consume_flags = [
                cleanup and consumers.get(src, 0) ==
                stage.inputs.count(src) and src not in outputs
                for src in stage.inputs]
            self._stage_live = {
                id(r) for k, v in data.items()
                if isinstance(v, PartStore) and k not in set(stage.inputs)
                for runs in v.values() for r in runs}
            with trace_stage(repr(stage), self.device):
                if isinstance(stage, GMap):
                    out = self.run_map(stage, ins)
                elif isinstance(stage, GReduce):
                    out = self.run_reduce(stage, ins)
                elif isinstance(stage, GSink):
                    out = self.run_sink(stage, ins)
                else:
                    raise TypeError(stage)
            data[stage.output] = out
            if cleanup:
                for src in set(stage.inputs):
                    consumers[src] -= stage.inputs.count(src)
                    if consumers.get(src, 0) <= 0:
                        victim = data.get(src)
                        # pass-through stages (identity/unkey/merges)
                        # share RUN objects between stores: free only
                        # runs not referenced by any other live source
                        live = {id(r)
                                for k, v in data.items()
                                if k != src and isinstance(v, PartStore)
                                for runs in v.values()
                                for r in runs}
                        self._free_store(victim, live)
                        data[src] = PartStore()
        rets = []
        for source in outputs:
            store = data[source]
            rets.append(self._collect(store))
        st = self.pool.stats()
        st["exchanged_rows"] = self.exchanged_rows
        self.stats = st
        if st["spilled_to_host_bytes"] or st["exchanged_rows"]:
            log.info("[device] run stats: %s", st)
        self.pool.cleanup()
        return rets

    def _free_run(self, run):
        self.pool.forget(run)
        run.drop()                  # joins in-flight spill/disk IO
        if run.on_disk:
            try:
                self.pool._os.unlink(run._disk)
            except OSError:
                pass
            run._disk = None

    def _consume_partition(self, ins, p, flags=None):
        """Free partition ``p`` of fully-consumed input stores (their
        last consumer is the running stage) once its reduce/join is
        done — bounds the tier churn of >pool jobs.  (The reference
        deletes intermediates only after ALL outputs are known,
        runner.py:207-228; per-partition granularity is what lets a
        120 GB job fit a 24 GB pool without re-evicting dead data.)"""
        if flags is None:
            flags = getattr(self, "_consume_flags", None)
        if not flags or len(flags) != len(ins):
            return
        live = getattr(self, "_stage_live", frozenset())
        seen = set()
        for st, f in zip(ins, flags):
            if not f or not isinstance(st, PartStore):
                continue
            for run in st.pop(p, []):
                if id(run) in live or id(run) in seen:
                    continue
                seen.add(id(run))
                self._free_run(run)

    def _free_store(self, store, live_ids=frozenset()):
        """Release a fully-consumed store's memory across all tiers
        (skipping runs still shared with live stores)."""
        if not isinstance(store, PartStore):
            return
        for part in list(store):
            kept = []
            for run in store[part]:
                if id(run) in live_ids:
                    kept.append(run)
                    continue
                self.pool.forget(run)
                run.drop()          # joins in-flight spill/disk IO
                if run.on_disk:
                    try:
                        self.pool._os.unlink(run._disk)
                    except OSError:
                        pass
                    run._disk = None
            if kept:
                store[part] = kept
            else:
                del store[part]

    # -- ingest / collect --------------------------------------------------

    def _ingest(self, inp):
        """Input -> partition store.  ColumnSource goes straight to device;
        host Datasets/Chunkers decode through the numeric encoder."""
        if isinstance(inp, TextSource):
            if self.world > 1:
                # per-rank newline-aligned slice of the corpus; every
                # rank computes the identical boundaries (same input),
                # so slices partition the text exactly
                t = inp.text
                n = t.shape[0]

                def _align(b):
                    # first byte AFTER the next newline at/past b
                    import numpy as np
                    if b <= 0 or b >= n:
                        return min(max(b, 0), n)
                    nl = np.flatnonzero(t[b - 1:] == ord("\n"))
                    return (b - 1 + int(nl[0]) + 1) if nl.size else n

                lo = _align(n * self.rank // self.world)
                hi = _align(n * (self.rank + 1) // self.world)
                return TextSource(t[lo:hi].copy())
            return inp
        if isinstance(inp, ColumnSource):
            keys = inp.keys.to(self.device)
            vals = inp.vals.to(self.device)
            if self.world > 1:
                # each rank keeps an equal slice of the input
                n = keys.numel()
                lo = n * self.rank // self.world
                hi = n * (self.rank + 1) // self.world
                keys, vals = keys[lo:hi], vals[lo:hi]
                st = self._partition(keys, vals, fkeys=inp.fkeys)
                st.str_table = inp.str_table
                return st
            if self.n_partitions == 1:
                st = self._partition(keys, vals, fkeys=inp.fkeys)
                st.str_table = inp.str_table
                return st
            # lazy ingest: batched unpartitioned runs as VIEWS of the
            # resident input — partitioning by the input keys is wasted
            # work when the first stage re-keys, and cloning would send
            # a full extra copy of the input through the spill tiers
            # (measured 2x tier traffic on a 120 GB job).  View runs
            # are NOT pool-admitted: the parent tensor is resident for
            # the stage regardless (ColumnSource holds it), so they are
            # never evicted and cost the pool nothing.
            store = PartStore(partitioned=False,
                              str_table=inp.str_table,
                              fkeys=inp.fkeys,
                              svals=_is_sv(inp.vals))
            store.vdtype = None if _is_sv(vals) else vals.dtype
            store[0] = []
            n = keys.numel()
            step = max(1, settings.gpu_batch_records)
            for lo in range(0, n, step):
                hi = min(lo + step, n)
                run = DeviceRun(keys[lo:hi], vals[lo:hi], sorted=False)
                store[0].append(run)
            return store
        # host dataset / chunker: stream records; numeric records become
        # columns, object records stay host-side (HostStore)
        records = self._host_records_of_input(inp)
        if self.world > 1:
            n = len(records)
            lo = n * self.rank // self.world
            hi = n * (self.rank + 1) // self.world
            records = records[lo:hi]
        return self._encode_or_host(records)

    @staticmethod
    def _host_records_of_input(inp):
        from ..dataset import Chunker, Dataset
        if isinstance(inp, Dataset):
            return list(inp.read())
        if isinstance(inp, Chunker):
            out = []
            for c in inp.chunks():
                out.extend(c.read())
            return out
        raise TypeError("GPU engine cannot ingest {!r}".format(inp))

    def _encode_records(self, records):
        import numpy as np
        if not records:
            return self._partition(
                torch.zeros(0, dtype=torch.int64),
                torch.zeros(0, dtype=torch.int64))
        ks = [k for k, _ in records]
        vs = [v for _, v in records]
        def _num(x):
            # bools stay Python objects: columnar i64 would decode True
            # as 1 (repr-visible divergence from the host engine)
            return isinstance(x, (int, float)) and not isinstance(x, bool)

        def _int(x):
            return isinstance(x, int) and not isinstance(x, bool)

        # host keyed-reducer convention: value = (key, scalar)
        keyed = all(
            isinstance(v, tuple) and len(v) == 2 and v[0] == k
            and (_num(v[1]) or isinstance(v[1], str))
            for (k, _), v in zip(records, vs))
        if keyed:
            vs = [v[1] for v in vs]
        if all(_int(v) for v in vs):
            vt = torch.from_numpy(np.asarray(vs, dtype=np.int64))
        elif all(_num(v) for v in vs):
            vt = torch.from_numpy(np.asarray(vs, dtype=np.float64))
        elif all(isinstance(v, str) for v in vs):
            from .strvals import StrVals
            vt = StrVals.from_strings(vs)
        else:
            raise TypeError(
                "device engine requires numeric or string values; use "
                "the host engine for object records")
        fkeys = False
        str_table = None
        if all(_int(k) for k in ks):
            kt = torch.from_numpy(np.asarray(ks, dtype=np.int64))
        elif all(_num(k) for k in ks):
            kt = _encode_f64_sortable(
                torch.from_numpy(np.asarray(ks, dtype=np.float64)))
            fkeys = True
        elif all(isinstance(k, str) for k in ks):
            # dictionary encoding: rank ids preserve lexicographic order
            str_table = tuple(sorted(set(ks)))
            rank = {t: i for i, t in enumerate(str_table)}
            kt = torch.from_numpy(
                np.fromiter((rank[k] for k in ks), dtype=np.int64,
                            count=len(ks)))
        else:
            raise TypeError(
                "device engine requires numeric or string keys; use the "
                "host engine for object records")
        store = self._partition(kt.to(self.device), vt.to(self.device),
                                keyed=keyed, fkeys=fkeys)
        store.str_table = str_table
        return store

    def _encode_or_host(self, records):
        if self.world > 1:
            return self._encode_records_world(records)
        try:
            return self._encode_records(records)
        except TypeError:
            return HostStore(records)

    def _encode_records_world(self, records):
        """world>1 encode: all ranks agree on ONE layout via a single
        all_gather of local metadata (key/value kinds, keyed flag,
        string table) before building any tensor.  A per-rank choice
        would (a) desync the exchange collectives when branches differ
        and (b) assign per-rank-incompatible string dictionary ids —
        the same id on two ranks decoding to different strings after
        the exchange."""
        import numpy as np
        import torch.distributed as dist

        def _num(x):
            return isinstance(x, (int, float)) and not isinstance(x, bool)

        def _int(x):
            return isinstance(x, int) and not isinstance(x, bool)

        ks = [k for k, _ in records]
        vs = [v for _, v in records]
        keyed = None
        if records:
            keyed = all(
                isinstance(v, tuple) and len(v) == 2 and v[0] == k
                and _num(v[1]) for k, v in zip(ks, vs))
        pv = [v[1] for v in vs] if keyed else vs

        def kind_of(xs, allow_str):
            if not xs:
                return "empty", None
            if all(_int(x) for x in xs):
                return "int", None
            if all(_num(x) for x in xs):
                return "float", None
            if allow_str and all(isinstance(x, str) for x in xs):
                return "str", tuple(sorted(set(xs)))
            return "obj", None

        kk, table = kind_of(ks, True)
        vk, _ = kind_of(pv, True)
        gathered = [None] * self.world
        dist.all_gather_object(gathered, (kk, vk, keyed, table))
        kks = {m[0] for m in gathered} - {"empty"}
        vks = {m[1] for m in gathered} - {"empty"}
        keyeds = {m[2] for m in gathered} - {None}
        if ("obj" in kks or "obj" in vks or len(keyeds) > 1
                or ("str" in kks and len(kks) > 1)
                or ("str" in vks and len(vks) > 1)):
            # no common columnar layout: every rank falls back to host
            # records (symmetric — no further collectives here; host
            # reduces exchange via _host_exchange)
            return HostStore(records)
        g_keyed = keyeds.pop() if keyeds else False
        vals = pv if g_keyed else vs
        if vks == {"str"}:
            from .strvals import StrVals
            vt = StrVals.from_strings(vals)
        else:
            v_np = np.int64 if vks in ({"int"}, set()) else np.float64
            vt = torch.from_numpy(np.asarray(vals, dtype=v_np))
        fkeys = False
        str_table = None
        if kks == {"str"}:
            str_table = tuple(sorted(
                set().union(*[set(m[3]) for m in gathered if m[3]])))
            ranks = {t: i for i, t in enumerate(str_table)}
            kt = torch.from_numpy(np.fromiter(
                (ranks[k] for k in ks), dtype=np.int64, count=len(ks)))
        elif kks in ({"int"}, set()):
            kt = torch.from_numpy(np.asarray(ks, dtype=np.int64))
        else:
            # float keys, or int+float mixed (ints beyond 2^53 would
            # lose precision here; such pipelines belong on the host
            # engine)
            kt = _encode_f64_sortable(
                torch.from_numpy(np.asarray(ks, dtype=np.float64)))
            fkeys = True
        store = self._partition(kt.to(self.device), vt.to(self.device),
                                keyed=bool(g_keyed), fkeys=fkeys)
        store.str_table = str_table
        return store

    def _collect(self, store):
        """Partition store -> one key-sorted ColumnDataset (the engine's
        MergeDataset analog: partitions are key-sorted, output is their
        merge)."""
        if isinstance(store, ColumnDataset):
            return store
        if isinstance(store, SinkStore):
            from ..dataset import cat_datasets
            return cat_datasets(store.datasets())
        if isinstance(store, HostStore):
            from ..dataset import MemoryDataset
            return MemoryDataset(sorted(store, key=lambda r: r[0]))
        if isinstance(store, TokenStore):
            return TokenColumnDataset(store, getattr(store, "keyed",
                                                     False))
        keyed = getattr(store, "keyed", False)
        fkeys = getattr(store, "fkeys", False)
        tbl = getattr(store, "str_table", None)
        ks, vs = [], []
        all_sorted = True
        for p in sorted(store):
            for run in store[p]:
                self.pool.touch(run, self.device)
                ks.append(run.keys)
                vs.append(run.vals)
                all_sorted = all_sorted and run.sorted
                self.pool.release(run)
        if not ks:
            z = torch.zeros(0, dtype=torch.int64)
            return ColumnDataset(z, z.clone(), keyed, fkeys, tbl)
        if all_sorted and len(ks) > 1:
            sk, perm = self.ops.merge_sorted_runs(ks, fkeys=fkeys)
            return ColumnDataset(sk, _cat_vals(vs)[perm], keyed, fkeys,
                                 tbl)
        keys = torch.cat(ks)
        vals = _cat_vals(vs)
        sk, sp = self._sort(keys, fkeys=fkeys)
        return ColumnDataset(sk, vals[sp.to(torch.int64)], keyed, fkeys,
                             tbl)

    # -- ordering ----------------------------------------------------------

    _SIGN = -(1 << 63)

    def _sort(self, keys, payload=None, fkeys=False):
        """Key sort in *host* ascending order: signed for i64 keys, float
        ascending for f64-encoded keys (their encoding is already
        unsigned-ascending).  The backend sort is unsigned (built for
        hashes), so int keys are biased through the sign bit."""
        if fkeys:
            return self.ops.sort_pairs(keys, payload)
        sk, sp = self.ops.sort_pairs(keys ^ self._SIGN, payload)
        return sk ^ self._SIGN, sp

    # -- partitioning ------------------------------------------------------

    def _partition(self, keys, vals, already_sorted=False, keyed=False,
                   fkeys=False):
        """Split columns into the partition store {p: [DeviceRun]} (K1+K2):
        partition ids, stable sort by id, slice contiguous segments."""
        P = self.n_partitions
        store = PartStore(keyed=keyed, fkeys=fkeys, svals=_is_sv(vals))
        store.vdtype = None if _is_sv(vals) else vals.dtype
        if keys.numel() == 0 and self.world == 1:
            return store
        if P == 1 and self.world == 1:
            # no routing needed: one resident partition
            if not already_sorted:
                run = DeviceRun(keys.contiguous(), vals.contiguous(),
                                sorted=False)
            else:
                run = DeviceRun(keys.contiguous(), vals.contiguous(),
                                sorted=True)
            store[0] = [run]
            self.pool.admit(run)
            return store
        pid = self.ops.partition_of(keys, P)
        order = torch.argsort(pid, stable=True)
        keys, vals, pid = keys[order], vals[order], pid[order]
        if self.world > 1:
            keys, vals, pid = self._exchange(keys, vals, pid)
            # received rows arrive grouped by sender, not by partition:
            # restore pid order for the contiguous slicing below
            order = torch.argsort(pid, stable=True)
            keys, vals, pid = keys[order], vals[order], pid[order]
        self._slice_into(store, keys, vals, pid,
                         already_sorted=already_sorted)
        return store

    def _count_batches(self, ins):
        """Upper bound on the (k, v) batches ``batches()`` yields,
        computed WITHOUT consuming anything — the local term of the
        rank-agreed chunk count for the exchange pipeline."""
        c = 0
        for store in ins:
            if isinstance(store, PartStore) and not store.partitioned:
                c += len(store.get(0, []))
            elif isinstance(store, PartStore):
                c += len(self._parts([store]))
        return c

    def _slice_into(self, store, keys, vals, pid, already_sorted=False):
        """Append contiguous partition slices of pid-ordered columns to
        ``store`` (this rank's owned partitions only at world > 1)."""
        P = self.n_partitions
        counts = torch.bincount(pid, minlength=P)
        offs = torch.cumsum(counts, 0) - counts
        counts_l = counts.tolist()
        offs_l = offs.tolist()
        for p in range(P):
            if self.world > 1 and p % self.world != self.rank:
                continue
            n = counts_l[p]
            if not n:
                continue
            o = offs_l[p]
            # clone, not view: spilled runs must release their HBM
            # (views pin the whole routed batch)
            k = keys[o:o + n].clone()
            v = vals[o:o + n].clone()
            # runs stay unsorted: consumers that need key order sort at
            # merge time, once per partition instead of once per
            # (run, partition) slice — thousands of tiny sorts otherwise
            run = DeviceRun(k, v, sorted=bool(already_sorted))
            store.setdefault(p, []).append(run)
            self.pool.admit(run)
        return store

    def _ensure_partitioned(self, store):
        """Route an unpartitioned store's runs to hash partitions, one
        run at a time (each run is bounded, so this streams through the
        pool).  At world > 1 the runs concatenate FIRST: run counts
        differ across ranks, and _partition exchanges — one collective
        per stage per rank is the invariant."""
        if not isinstance(store, PartStore) or store.partitioned:
            return store
        if self.world > 1:
            ks, vs = [], []
            for run in store.get(0, []):
                self.pool.touch(run, self.device)
                ks.append(run.keys)
                vs.append(run.vals)
                self.pool.release(run)
            k = torch.cat(ks) if ks else torch.zeros(
                0, dtype=torch.int64, device=self.device)
            if vs:
                v = _cat_vals(vs)
            elif getattr(store, "svals", False):
                # empty rank: the exchange layout must still match the
                # other ranks' (var-len blob wire shape, value dtype)
                from .strvals import StrVals
                v = StrVals.empty(self.device)
            else:
                v = torch.zeros(
                    0,
                    dtype=getattr(store, "vdtype", None) or torch.int64,
                    device=self.device)
            out = self._partition(k, v, keyed=store.keyed,
                                  fkeys=store.fkeys)
            out.str_table = getattr(store, "str_table", None)
            return out
        out = None
        for run in store.get(0, []):
            self.pool.touch(run, self.device)
            k, v = run.keys, run.vals
            self.pool.release(run)
            part = self._partition(k, v, keyed=store.keyed,
                                   fkeys=store.fkeys)
            if out is None:
                out = part
            else:
                for q, runs in part.items():
                    out.setdefault(q, []).extend(runs)
        if out is None:
            out = PartStore(keyed=store.keyed, fkeys=store.fkeys,
                            svals=getattr(store, "svals", False))
        out.str_table = getattr(store, "str_table", None)
        return out

    def _exchange(self, keys, vals, pid):
        """RCCL all-to-all: route rows to the partition's owning rank
        (p % world); returns this rank's rows."""
        from ..parallel.shuffle import (exchange_columns,
                                        exchange_columns_varlen)
        self.exchanged_rows += keys.numel()
        if _is_sv(vals):
            return exchange_columns_varlen(keys, vals, pid, self.world)
        return exchange_columns(keys, vals, pid, self.world)

    def _merged_partition(self, stores, p, need_sorted=True):
        """All runs of partition p across input stores, merged
        key-sorted (``need_sorted=False`` skips the sort for consumers
        that re-key anyway, e.g. the kv map)."""
        fkeys = any(getattr(s, "fkeys", False) for s in stores)
        ks, vs = [], []
        all_sorted = True
        for store in stores:
            for run in store.get(p, []):
                self.pool.touch(run, self.device)
                ks.append(run.keys)
                vs.append(run.vals)
                all_sorted = all_sorted and run.sorted
                self.pool.release(run)
        if not ks:
            return None, None
        if len(ks) == 1 and (all_sorted or not need_sorted):
            return ks[0], vs[0]
        if not need_sorted:
            return torch.cat(ks), _cat_vals(vs)
        if all_sorted and len(ks) > 1:
            # K4: merge-path k-way merge of already-sorted runs — one
            # log2(R)-deep pass tree instead of a full radix re-sort
            # (reference analog: heapq.merge, dataset.py:567-588)
            mk, perm = self.ops.merge_sorted_runs(ks, fkeys=fkeys)
            return mk, _cat_vals(vs)[perm]
        keys = torch.cat(ks)
        vals = _cat_vals(vs)
        sk, sp = self._sort(keys, fkeys=fkeys)
        return sk, vals[sp.to(torch.int64)]

    def _all_rows(self, store):
        """All (keys, vals) rows of a store, concatenated on device."""
        ks, vs = [], []
        for p in sorted(store):
            k, v = self._merged_partition([store], p, need_sorted=False)
            if k is None:
                continue
            ks.append(k)
            vs.append(v)
        if not ks:
            z = torch.zeros(0, dtype=torch.int64, device=self.device)
            return z, z.clone()
        return torch.cat(ks), _cat_vals(vs)

    def _parts(self, stores):
        ps = set()
        for s in stores:
            ps.update(s.keys())
        return sorted(ps)

    # -- map stage ---------------------------------------------------------

    def run_map(self, stage, ins):
        spec = stage.options.get("device_map")
        if spec is None or any(isinstance(s, HostStore) for s in ins):
            return self._host_map(stage, ins)
        kind = spec[0]
        if kind == "kv":
            if any(getattr(s, "keyed", False) for s in ins):
                # tuple-valued records (keyed convention): column funcs
                # don't apply — run the stage's Python mapper instead
                return self._host_map(stage, ins)
            # emit (keyfn(v), valfn(v)) per record — the group_by/count map
            _kind, keyf, valf = spec
            if keyf == "identity" and any(_store_has_sv(s) for s in ins):
                # keying by a var-len VALUE needs a dictionary encode;
                # host path builds it (string-key ingest re-enters the
                # device path downstream)
                return self._host_map(stage, ins)
            out = None

            def kv_batch(keys, vals):
                nk = self._apply_colfunc(keyf, keys, vals)
                nv = self._apply_colfunc(valf, keys, vals)
                fkeys = nk.dtype == torch.float64
                if fkeys:
                    nk = _encode_f64_sortable(nk)
                return self._partition(nk, nv, fkeys=fkeys)

            def fold(out, part):
                if out is None:
                    return part
                for q, runs in part.items():
                    out.setdefault(q, []).extend(runs)
                return out

            def batches():
                # stream-consume: a batch's source runs are freed as
                # soon as the kv transform's output exists (this stage
                # is their last consumer), so the routed copy does not
                # coexist with the whole unrouted input on >pool jobs
                flags = getattr(self, "_consume_flags", None)
                if not flags or len(flags) != len(ins):
                    flags = [False] * len(ins)
                live = getattr(self, "_stage_live", frozenset())
                for store, f in zip(ins, flags):
                    if isinstance(store, PartStore) \
                            and not store.partitioned:
                        # record-wise op: stream run by run
                        for run in store.get(0, []):
                            self.pool.touch(run, self.device)
                            k, v = run.keys, run.vals
                            self.pool.release(run)
                            yield k, v
                            if f and id(run) not in live:
                                self._free_run(run)
                        if f:
                            store.pop(0, None)
                    else:
                        for p in self._parts([store]):
                            keys, vals = self._merged_partition(
                                [store], p, need_sorted=False)
                            if keys is None:
                                continue
                            yield keys, vals
                            self._consume_partition([store], p,
                                                    flags=[f])

            if self.world > 1:
                # CHUNKED exchange pipeline: ranks agree on the chunk
                # COUNT (one max-reduce; the collective sequence per
                # stage stays identical on every rank), then per chunk:
                # column funcs + partition routing on the compute
                # stream, the data all-to-all on RCCL's comm stream.
                # Routing chunk i+1 is issued while chunk i's exchange
                # is in flight (only the small counts collective syncs
                # the host); received chunks slice into runs at the
                # end.  (SURVEY §2.3 overlap; gloo executes the same
                # sequence synchronously.)
                import torch.distributed as dist
                from ..parallel.shuffle import (exchange_columns,
                                                exchange_columns_varlen)
                t = torch.tensor([self._count_batches(ins)])
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                C = int(t.item())
                # rank-deterministic empty-chunk layout (an empty rank
                # must match the others' collective dtypes)
                vd = next((getattr(s, "vdtype", None) for s in ins
                           if getattr(s, "vdtype", None) is not None),
                          torch.int64)
                in_sv = any(_store_has_sv(s) for s in ins)
                # fkeys derives from world-agreed metadata, NOT from an
                # observed batch: a rank with zero local batches must
                # still treat RECEIVED keys as f64-encoded when the
                # other ranks encode
                fkeys = (keyf == "identity" and vd == torch.float64) \
                    or (keyf == "key"
                        and any(getattr(s, "fkeys", False)
                                for s in ins))
                it = batches()
                parts = []
                for _i in range(C):
                    try:
                        k, v = next(it)
                    except StopIteration:
                        k = v = None
                    if k is None:
                        nk = torch.zeros(0, dtype=torch.int64,
                                         device=self.device)
                        if valf == "identity" and in_sv:
                            from .strvals import StrVals
                            nv = StrVals.empty(self.device)
                        else:
                            nv = torch.zeros(
                                0,
                                dtype=(vd if valf == "identity"
                                       else torch.int64),
                                device=self.device)
                    else:
                        nk = self._apply_colfunc(keyf, k, v)
                        nv = self._apply_colfunc(valf, k, v)
                        assert (nk.dtype == torch.float64) == fkeys, \
                            "kv key dtype disagrees with store metadata"
                    if fkeys:
                        nk = _encode_f64_sortable(nk)
                    pid = self.ops.partition_of(nk, self.n_partitions)
                    order = torch.argsort(pid, stable=True)
                    nk, nv, pid = nk[order], nv[order], pid[order]
                    self.exchanged_rows += nk.numel()
                    if _is_sv(nv):
                        parts.append(exchange_columns_varlen(
                            nk, nv, pid, self.world))
                    else:
                        parts.append(exchange_columns(
                            nk, nv, pid, self.world))
                out = PartStore(fkeys=bool(fkeys))
                rk = torch.cat([e[0] for e in parts]) if parts else \
                    torch.zeros(0, dtype=torch.int64, device=self.device)
                rv = _cat_vals([e[1] for e in parts]) if parts else \
                    torch.zeros(0, dtype=torch.int64, device=self.device)
                rp = torch.cat([e[2] for e in parts]) if parts else \
                    torch.zeros(0, dtype=torch.int64, device=self.device)
                order = torch.argsort(rp, stable=True)
                self._slice_into(out, rk[order], rv[order], rp[order])
                out.svals = _is_sv(rv)
                out.vdtype = None if _is_sv(rv) else rv.dtype
                return out
            for k, v in batches():
                out = fold(out, kv_batch(k, v))
            return out if out is not None else PartStore()
        if kind == "identity":
            return self._merge_stores(ins)
        if kind == "unkey":
            # strip the keyed-reducer value convention: values become the
            # bare aggregates (the host stage extracts x[1])
            merged = self._merge_stores(ins)
            if isinstance(merged, (HostStore, TokenStore)) \
                    or not isinstance(merged, PartStore):
                return self._host_map(stage, ins)
            out = PartStore(keyed=False, fkeys=merged.fkeys,
                            partitioned=merged.partitioned,
                            str_table=merged.str_table,
                            svals=getattr(merged, "svals", False))
            for q, runs in merged.items():
                out[q] = runs
            return out
        if kind == "len_local":
            # row count from run metadata: no data is read at all
            total = 0
            ok = True
            for store in ins:
                if isinstance(store, PartStore):
                    for runs in store.values():
                        for r in runs:
                            total += r.n
                else:
                    ok = False
            if not ok:
                return self._host_map(stage, ins)
            if self.world > 1:
                keys = torch.ones(1, dtype=torch.int64,
                                  device=self.device)
                vals = torch.tensor([total], dtype=torch.int64,
                                    device=self.device)
                return self._partition(keys, vals)
            if total == 0:
                return PartStore()
            store = PartStore()
            run = DeviceRun(
                torch.ones(1, dtype=torch.int64, device=self.device),
                torch.tensor([total], dtype=torch.int64,
                             device=self.device), sorted=True)
            store[0] = [run]
            self.pool.admit(run)
            return store
        if kind == "text_df":
            src = ins[0]
            if not isinstance(src, TextSource) or \
                    self.device.type != "cuda":
                return self._host_map(stage, ins)
            from .tfidf import TfidfEngine
            if src.text_t is not None:
                text = src.text_t.to(self.device)
            else:
                text = torch.from_numpy(src.text).to(self.device)
            eng = TfidfEngine(self.device)
            eng.reset()
            n = text.numel()
            cb = (1 << 30)               # 1 GiB chunks, newline-aligned
            bounds = [0]
            while bounds[-1] < n:
                e = min(bounds[-1] + cb, n)
                if e < n:
                    seg = text[e - 1:min(e + (1 << 16), n)]
                    nl = torch.nonzero(seg == ord("\n")).flatten()
                    e = (e - 1 + int(nl[0].item()) + 1) if nl.numel() \
                        else n
                bounds.append(e)
            for s0, e0 in zip(bounds, bounds[1:]):
                eng.count_chunk(text[s0:e0].contiguous(), pos_base=s0)
            keys, df = eng.extract()
            if self.world > 1:
                # ONE exchange of (key, df, token-bytes) partials routed
                # by key % world; each rank rebuilds its owned shard
                # (same pattern as bench.py's explicit pipeline)
                from ..parallel.shuffle import (all_reduce_scalar,
                                                exchange_keyed_payload)
                blob, lens = eng.token_strings_dev(keys, text)
                rk, rdf, rblob, rlens = exchange_keyed_payload(
                    keys, df, blob, lens)
                eng.merge_exchanged(rk, rdf, rblob, rlens)
                keys, df = eng.extract()
                eng.n_docs = all_reduce_scalar(eng.n_docs,
                                               device=self.device)
                # received token strings live in the exchanged blob
                text = rblob
            return TokenStore(eng, keys, df, text)
        if kind == "cross":
            # K9 broadcast cross join: ins[0] = streamed (outer) side
            # whose keys the output carries, ins[1] = broadcast side
            # (gathered across ranks like the host supplemental path).
            opn = spec[1]
            if len(ins) != 2 \
                    or any(not isinstance(s, PartStore) for s in ins) \
                    or any(getattr(s, "keyed", False) for s in ins) \
                    or any(_store_has_sv(s) for s in ins):
                return self._host_map(stage, ins)
            other, me = ins
            ko, vo = self._all_rows(other)
            km, vm = self._all_rows(me)
            if self.world > 1:
                from ..parallel.shuffle import gather_columns
                km, vm = gather_columns(km, vm, device=self.device)
            n_o, m = ko.numel(), km.numel()
            out_k = torch.repeat_interleave(ko, m) if m else \
                torch.zeros(0, dtype=torch.int64, device=self.device)
            if n_o and m:
                a = vm.repeat(n_o)                   # me value (v2)
                b = torch.repeat_interleave(vo, m)   # other value (v1)
                if a.dtype != b.dtype:
                    dt = torch.promote_types(a.dtype, b.dtype)
                    a, b = a.to(dt), b.to(dt)
                out_v = {"add": lambda: a + b,
                         "mul": lambda: a * b,
                         "min": lambda: torch.minimum(a, b),
                         "max": lambda: torch.maximum(a, b)}[opn]()
            else:
                out_v = torch.zeros(0, dtype=torch.int64,
                                    device=self.device)
            st = self._partition(out_k, out_v,
                                 fkeys=getattr(other, "fkeys", False))
            st.str_table = getattr(other, "str_table", None)
            return st
        if kind == "cross_set":
            # K9 broadcast-set: fold the broadcast side (ins[1]) to one
            # scalar, apply cross(v_streamed, scalar) over ins[0]
            opn, aggn = spec[1], spec[2]
            if len(ins) != 2 \
                    or any(not isinstance(s, PartStore) for s in ins) \
                    or any(getattr(s, "keyed", False) for s in ins) \
                    or any(_store_has_sv(s) for s in ins):
                return self._host_map(stage, ins)
            other, me = ins
            ko, vo = self._all_rows(other)
            km, vm = self._all_rows(me)
            if self.world > 1:
                from ..parallel.shuffle import gather_columns
                km, vm = gather_columns(km, vm, device=self.device)
            if vm.numel() == 0:
                # host semantics: sum() of nothing is 0, min/max raise
                return self._host_map(stage, ins)
            scalar = {"sum": vm.sum, "min": vm.min, "max": vm.max}[aggn]()
            a, b = vo, scalar
            if a.dtype != b.dtype:
                dt = torch.promote_types(a.dtype, b.dtype)
                a, b = a.to(dt), b.to(dt)
            out_v = {"add": lambda: a + b,
                     "mul": lambda: a * b,
                     "min": lambda: torch.minimum(a, b),
                     "max": lambda: torch.maximum(a, b)}[opn]()
            st = self._partition(ko, out_v,
                                 fkeys=getattr(other, "fkeys", False))
            st.str_table = getattr(other, "str_table", None)
            return st
        if kind == "topk_local":
            # per-partition top-k candidates by value (K11); all
            # candidates meet in partition 0 for the global pass
            if any(getattr(s, "keyed", False) for s in ins) \
                    or any(_store_has_sv(s) for s in ins):
                return self._host_map(stage, ins)
            K = spec[1]
            cand_k, cand_v = [], []
            fkeys = False

            def topk_batch(vals):
                nonlocal fkeys
                if vals.dtype == torch.float64:
                    enc = _encode_f64_sortable(vals)
                    fkeys = True
                else:
                    enc = vals
                sk, sp = self._sort(enc, fkeys=vals.dtype
                                    == torch.float64)
                top = min(K, sk.numel())
                cand_k.append(sk[-top:])
                cand_v.append(vals[sp.to(torch.int64)[-top:]])

            for store in ins:
                if isinstance(store, PartStore) \
                        and not store.partitioned:
                    for run in store.get(0, []):
                        self.pool.touch(run, self.device)
                        v = run.vals
                        self.pool.release(run)
                        topk_batch(v)
                    continue
                for p in self._parts([store]):
                    keys, vals = self._merged_partition(
                        [store], p, need_sorted=False)
                    if keys is None:
                        continue
                    topk_batch(vals)
            store = PartStore(fkeys=fkeys)
            if not cand_k:
                if self.world == 1:
                    return store
                # candidate-less rank (input shard smaller than the
                # world) must still join the exchange with the agreed
                # layout, or the collective sequence desyncs
                vd = next((getattr(s, "vdtype", None) for s in ins
                           if getattr(s, "vdtype", None) is not None),
                          torch.int64)
                fkeys = vd == torch.float64
                store.fkeys = fkeys
                ck = torch.zeros(0, dtype=torch.int64,
                                 device=self.device)
                cv = torch.zeros(0, dtype=vd, device=self.device)
            else:
                ck = torch.cat(cand_k)
                cv = torch.cat(cand_v)
            if self.world > 1:
                zeros = torch.zeros_like(ck)
                ck, cv, _ = self._exchange(ck, cv, zeros)
                if self.rank != 0 or ck.numel() == 0:
                    return store
            sk, sp = self._sort(ck, fkeys=fkeys)
            run = DeviceRun(sk, cv[sp.to(torch.int64)], sorted=True)
            store[0] = [run]
            self.pool.admit(run)
            return store
        raise ValueError("unknown device_map spec {!r}".format(spec))

    def _apply_colfunc(self, name, keys, vals):
        if name == "identity":
            return vals
        if name == "one":
            return torch.ones_like(keys)
        if name == "key":
            return keys
        raise ValueError("unknown column func {!r}".format(name))

    def _unify_str_stores(self, stores):
        """Cross-encode string keys (ROADMAP 2's remap): rank ids from
        different dictionaries are incompatible, so remap every run's
        key column through a device lut into the union table.  The lut
        is monotone (sorted table -> sorted union), so sorted runs stay
        sorted; hash ROUTING changes, so remapped stores come back
        unpartitioned and re-route in _ensure_partitioned.  Returns the
        unified store list, or None when any input has no dictionary
        (numeric/host/token stores: caller falls back to host records)."""
        tables = [getattr(s, "str_table", None) for s in stores]
        if not all(isinstance(s, PartStore) and t is not None
                   for s, t in zip(stores, tables)):
            return None
        merged = tuple(sorted(set().union(*map(set, tables))))
        index = {t: i for i, t in enumerate(merged)}
        out = []
        for s, t in zip(stores, tables):
            if t == merged:
                out.append(s)
                continue
            lut = torch.tensor([index[x] for x in t], dtype=torch.int64,
                               device=self.device)
            ns = PartStore(keyed=getattr(s, "keyed", False),
                           partitioned=False, str_table=merged,
                           svals=getattr(s, "svals", False))
            ns.vdtype = getattr(s, "vdtype", None)
            ns[0] = []
            for p in sorted(s):
                for run in s[p]:
                    self.pool.touch(run, self.device)
                    nr = DeviceRun(lut[run.keys], run.vals.clone(),
                                   sorted=run.sorted)
                    self.pool.release(run)
                    ns[0].append(nr)
                    self.pool.admit(nr)
            out.append(ns)
        return out

    def _unify_fkeys_stores(self, stores):
        """Mixed float-keyed and int-keyed inputs: the f64 side's keys are
        order-preserving encodings (encode_f64_sortable) while the int
        side's are raw i64 bit patterns, so comparisons AND hash routing
        would silently miss (1 != encode(1.0)).  Re-encode the int side
        through float64 (host semantics: 1 == 1.0).  Ints beyond 2^53
        lose precision here — the same contract as the mixed int+float
        record encode (_encode_records_world); such pipelines belong on
        the host engine.  Re-encoded stores come back unpartitioned
        (routing changed) and re-route in _ensure_partitioned."""
        flags = [bool(getattr(s, "fkeys", False)) for s in stores]
        if all(flags) or not any(flags):
            return stores
        out = []
        for s, f in zip(stores, flags):
            if f or not isinstance(s, PartStore):
                out.append(s)
                continue
            ns = PartStore(keyed=getattr(s, "keyed", False), fkeys=True,
                           partitioned=False,
                           str_table=getattr(s, "str_table", None),
                           svals=getattr(s, "svals", False))
            ns.vdtype = getattr(s, "vdtype", None)
            ns[0] = []
            for p in sorted(s):
                for run in s[p]:
                    self.pool.touch(run, self.device)
                    # monotone in signed i64 order, so sorted runs stay
                    # sorted
                    nk = _encode_f64_sortable(run.keys.to(torch.float64))
                    nr = DeviceRun(nk, run.vals.clone(), sorted=run.sorted)
                    self.pool.release(run)
                    ns[0].append(nr)
                    self.pool.admit(nr)
            out.append(ns)
        return out

    def _merge_stores(self, stores):
        if len(stores) == 1:
            return stores[0]
        if any(not isinstance(s, PartStore) for s in stores):
            # HostStore / TokenStore / TextSource in the mix: combine as
            # host records (only PartStores share the run layout)
            out = HostStore()
            for s in stores:
                out.extend(self._decode_store(s))
            return out
        if len(stores) > 1 and any(
                getattr(s, "str_table", None) is not None
                for s in stores):
            uni = self._unify_str_stores(stores)
            if uni is None:
                # dictionary + dictionary-less mix: combine as host
                # records (correct, slower)
                out = HostStore()
                for s in stores:
                    out.extend(self._decode_store(s))
                return out
            stores = uni
        sv_flags = [_store_has_sv(s) for s in stores]
        nonempty = [s for s in stores
                    if isinstance(s, PartStore) and len(s)]
        if any(sv_flags) and not all(
                _store_has_sv(s) for s in nonempty):
            # var-len + numeric value mix: no common column layout —
            # combine as host records (correct, slower)
            out = HostStore()
            for s in stores:
                out.extend(self._decode_store(s))
            return out
        if all(isinstance(s, PartStore) for s in stores):
            stores = self._unify_fkeys_stores(stores)
        flags = [getattr(s, "partitioned", True) for s in stores]
        if not all(flags) and any(flags):
            # mixing hashed and unrouted partition-0 runs would corrupt
            # co-location: route everything first
            stores = [self._ensure_partitioned(s) for s in stores]
            flags = [getattr(s, "partitioned", True) for s in stores]
        out = PartStore(
            keyed=any(getattr(s, "keyed", False) for s in stores),
            fkeys=any(getattr(s, "fkeys", False) for s in stores),
            svals=any(getattr(s, "svals", False) for s in stores),
            partitioned=all(flags),
            str_table=next((getattr(s, "str_table", None)
                            for s in stores
                            if getattr(s, "str_table", None) is not None),
                           None))
        out.vdtype = next((getattr(s, "vdtype", None) for s in stores
                           if getattr(s, "vdtype", None) is not None),
                          None)
        for s in stores:
            for p, runs in s.items():
                out.setdefault(p, []).extend(runs)
        return out

    # -- reduce stage ------------------------------------------------------

    def run_reduce(self, stage, ins):
        spec = stage.options.get("device_reduce")
        uni_table = None
        if len(ins) > 1 and any(
                getattr(s, "str_table", None) is not None for s in ins):
            uni = self._unify_str_stores(ins)
            if uni is None:
                return self._host_reduce(stage, ins)
            ins = uni
            uni_table = ins[0].str_table
        if len(ins) > 1 and all(isinstance(s, PartStore) for s in ins):
            ins = self._unify_fkeys_stores(ins)
        ins = [self._ensure_partitioned(s) for s in ins]
        if len(ins) == 1 and isinstance(ins[0], TokenStore) \
                and spec == ("sum",):
            ins[0].keyed = True        # keyed-reducer output convention
            return ins[0]
        if spec is None or any(getattr(s, "keyed", False) for s in ins) \
                or any(isinstance(s, HostStore) for s in ins):
            return self._host_reduce(stage, ins)
        kind = spec[0]
        if any(_store_has_sv(s) for s in ins):
            # var-len values: first/join(left|right) stay on device
            # (gather-only ops); arithmetic folds go to host records
            sv_ok = kind == "first" or (
                kind == "join" and stage.options.get(
                    "device_join_pair") in ("left", "right"))
            if not sv_ok:
                return self._host_reduce(stage, ins)
            if kind != "join" and len(ins) > 1:
                svs = [_store_has_sv(s) for s in ins
                       if isinstance(s, PartStore) and len(s)]
                if any(svs) and not all(svs):
                    # var-len + numeric mix can't share one value column
                    return self._host_reduce(stage, ins)
        in_fkeys = any(getattr(s, "fkeys", False) for s in ins)
        out = PartStore(keyed=True, fkeys=in_fkeys,
                        svals=any(_store_has_sv(s) for s in ins),
                        str_table=getattr(ins[0], "str_table", None)
                        if len(ins) == 1 else uni_table)
        if kind in ("sum", "min", "max"):
            parts = self._parts(ins)
            for i, p in enumerate(parts):
                self._prefetch_partition(ins, parts, i + 1)
                self._wait_prefetch()
                uk, agg = self._reduce_partition(ins, p, kind)
                self._consume_partition(ins, p)
                if uk is None:
                    continue
                run = DeviceRun(uk, agg, sorted=True)
                out.setdefault(p, []).append(run)
                self.pool.admit(run)
            return out
        if kind == "mean":
            for p in self._parts(ins):
                uk, sm = self._reduce_partition(
                    ins, p, "sum", vt=lambda v: v.to(torch.float64))
                if uk is None:
                    continue
                _uk, c = self._reduce_partition(
                    ins, p, "sum",
                    vt=lambda v: torch.ones_like(v,
                                                 dtype=torch.float64))
                self._consume_partition(ins, p)
                run = DeviceRun(uk, sm / c, sorted=True)
                out.setdefault(p, []).append(run)
                self.pool.admit(run)
            return out
        if kind == "first":
            # first value per key: stable sorts keep insertion order
            # within equal keys, so the segment head is the first seen
            for p in self._parts(ins):
                uk, fv = self._reduce_partition(ins, p, "first")
                self._consume_partition(ins, p)
                if uk is None:
                    continue
                run = DeviceRun(uk, fv, sorted=True)
                out.setdefault(p, []).append(run)
                self.pool.admit(run)
            return out
        if kind == "topk_global":
            K = spec[1]
            ks, vs = [], []
            for p in self._parts(ins):
                keys, vals = self._merged_partition(ins, p)
                if keys is None:
                    continue
                ks.append(keys)
                vs.append(vals)
            out.keyed = False
            if not ks:
                return out
            keys = torch.cat(ks)
            vals = torch.cat(vs)
            sk, sp = self._sort(keys, fkeys=in_fkeys)
            top = min(K, sk.numel())
            run = DeviceRun(sk[-top:].contiguous(),
                            vals[sp.to(torch.int64)[-top:]].contiguous(),
                            sorted=True)
            out[0] = [run]
            self.pool.admit(run)
            return out
        if kind == "join":
            how = spec[1]
            assert len(ins) == 2, "join takes two inputs"
            res = self._device_join(ins[0], ins[1], how, stage)
            res.str_table = uni_table
            return res
        raise ValueError("unknown device_reduce spec {!r}".format(spec))

    def _prefetch_partition(self, ins, parts, i):
        """Kick off async page-in of partition parts[i] on the side
        stream (no-op without a CUDA device)."""
        if self._side_stream is None or i >= len(parts):
            return
        nxt = parts[i]
        runs = [r for store in ins
                if isinstance(store, PartStore)
                for r in store.get(nxt, [])]
        self.pool.prefetch(runs, self.device, self._side_stream)

    def _wait_prefetch(self):
        if self._side_stream is not None:
            torch.cuda.current_stream(self.device).wait_stream(
                self._side_stream)

    def _first_sorted(self, keys, vals):
        """(unique_keys, first value per key) over a key-sorted column;
        stable sorts preserve insertion order within equal keys."""
        _seg, uk = self.ops.segment_ids(keys)
        flags = torch.ones_like(keys)
        flags[1:] = (keys[1:] != keys[:-1]).to(torch.int64)
        first_idx = torch.nonzero(flags.bool()).flatten()
        return uk, vals[first_idx]

    def _reduce_partition(self, ins, p, kind, vt=None):
        """One partition's reduce (kind: sum/min/max/first; ``vt``
        transforms the value column first).  A skewed partition whose
        runs exceed half the pool is reduced run-by-run and the partial
        aggregates re-reduced (associativity; "first" re-reduces in run
        order under stable sorts) — bounded memory instead of a giant
        merge (SURVEY §7 "skewed keys")."""
        def one(k_sorted, v):
            if kind == "first":
                return self._first_sorted(k_sorted, v)
            return self.ops.seg_reduce_sorted(k_sorted, v, kind)

        runs = [r for store in ins for r in store.get(p, [])]
        if not runs:
            return None, None
        fkeys = any(getattr(st, "fkeys", False) for st in ins)
        total = sum(r.nbytes for r in runs)
        if total <= self.pool.capacity // 2 or len(runs) == 1:
            keys, vals = self._merged_partition(ins, p)
            return one(keys, vt(vals) if vt else vals)
        pk, pv = [], []
        for run in runs:
            self.pool.touch(run, self.device)
            k, v = run.keys, run.vals
            self.pool.release(run)
            if not run.sorted:
                k, sp = self._sort(k, fkeys=fkeys)
                v = v[sp.to(torch.int64)]
            uk, agg = one(k, vt(v) if vt else v)
            pk.append(uk)
            pv.append(agg)
        keys = torch.cat(pk)
        vals = _cat_vals(pv)
        sk, sp = self._sort(keys, fkeys=fkeys)
        return one(sk, vals[sp.to(torch.int64)])

    def _device_join(self, left, right, how, stage):
        """Per-partition device hash join (K8); emits matched value pairs.
        The DSL's aggregate(left_vals, right_vals) runs per pair on the
        host only when not a recognized columnar op; the default device
        aggregate emits (key, (lv, rv)) as two columns folded into one
        f64/i64 value via the stage's pair op, or keeps lv when
        aggregate is 'left'."""
        pair_op = stage.options.get("device_join_pair", "pair_host")
        out = PartStore(
            keyed=True,
            fkeys=getattr(left, "fkeys", False)
            or getattr(right, "fkeys", False),
            svals=(_store_has_sv(left) if pair_op == "left" else
                   _store_has_sv(right) if pair_op == "right" else
                   False))
        cap = int(os.environ.get("DAMPR_JOIN_PROBE_ROWS",
                                 settings.gpu_join_probe_rows))
        for p in self._parts([left, right]):
            # probe keys are CLUSTERED by their low bits (hash slot =
            # key & mask) rather than fully sorted: probes walk
            # L2-resident table regions (measured 3.5x over random
            # probing round 1) and the partial sort costs 2 radix
            # passes instead of up to 8
            lk, lv = self._merged_partition([left], p,
                                            need_sorted=False)
            rk, rv = self._merged_partition([right], p,
                                            need_sorted=False)
            if lk is None and rk is None:
                continue
            if lk is None:
                lk = torch.zeros(0, dtype=torch.int64, device=self.device)
                lv = torch.zeros(0, dtype=torch.int64, device=self.device)
            if rk is None:
                rk = torch.zeros(0, dtype=torch.int64, device=self.device)
                rv = torch.zeros(0, dtype=torch.int64, device=self.device)
            # skewed-join guard (ROADMAP 7): the table is built on the
            # RIGHT side, probes stream the LEFT.  Inner joins are
            # symmetric, so if the build side is the oversized one, swap
            # so the big side is probed in batches instead of blowing
            # out the table build.
            swap = (how == "inner" and rk.numel() > cap
                    and rk.numel() > lk.numel())
            if swap:
                lk, lv, rk, rv = rk, rv, lk, lv
            # probe through a low-bit-clustered VIEW of the keys; only
            # the (small) emitted index set maps back through the
            # permutation — the value column is never re-gathered
            perm = self.ops.probe_order(lk)
            probe_k = lk[perm] if perm is not None else lk
            # build the table ONCE; the skew guard probes in batches
            table = self.ops.hash_join_build(rk)
            # inner/left matches of a probe row are independent of every
            # other probe row, so probing distributes over contiguous
            # batches; full-outer tracks unmatched RIGHT rows across the
            # whole probe and must run in one piece.
            n = probe_k.numel()
            step = n if (how == "outer" or n <= cap) else cap
            for a in range(0, max(n, 1), max(step, 1)):
                b = min(n, a + max(step, 1))
                li, ri = self.ops.hash_join(probe_k[a:b], rk, how,
                                            table=table)
                li = torch.where(li >= 0, li + a, li)
                if perm is not None:
                    li = torch.where(
                        li >= 0, perm[torch.clamp(li, min=0)], li)
                if swap:
                    li, ri = ri, li
                    klk, krk, klv, krv = rk, lk, rv, lv
                else:
                    klk, krk, klv, krv = lk, rk, lv, rv
                keys = torch.where(li >= 0, klk[torch.clamp(li, min=0)],
                                   krk[torch.clamp(ri, min=0)])
                valid_l = li >= 0
                valid_r = ri >= 0
                if pair_op == "left":
                    merged = self._side_val(klv, li, valid_l)
                elif pair_op == "right":
                    merged = self._side_val(krv, ri, valid_r)
                else:
                    lvm = self._side_val(klv, li, valid_l)
                    rvm = self._side_val(krv, ri, valid_r)
                    merged = self._apply_pair_op(pair_op, lvm, rvm,
                                                 valid_l, valid_r)
                if keys.numel() == 0 and n > 0:
                    continue
                # emitted unsorted: consumers (collect/reduce) sort the
                # whole output once instead of per-batch sort + merge
                run = DeviceRun(keys.contiguous(), merged.contiguous()
                                if not _is_sv(merged) else merged,
                                sorted=False)
                out.setdefault(p, []).append(run)
                self.pool.admit(run)
            self._consume_partition([left, right], p)
        return out

    @staticmethod
    def _side_val(vcol, idx, valid):
        """One join side's matched values by row index; unmatched rows
        get the dtype's zero (numeric) or empty bytes (var-len)."""
        if _is_sv(vcol):
            from .strvals import StrVals
            if bool(valid.all()):
                return vcol.gather(idx)
            # append an empty row as the unmatched target
            ext = StrVals(vcol.blob,
                          torch.cat([vcol.offs, vcol.offs[-1:]]))
            nrow = vcol.numel()
            return ext.gather(
                torch.where(valid, idx, torch.full_like(idx, nrow)))
        if vcol.numel() == 0:           # empty side: all rows unmatched
            return torch.zeros(idx.numel(), dtype=vcol.dtype,
                               device=idx.device)
        return torch.where(valid, vcol[torch.clamp(idx, min=0)],
                           torch.zeros_like(idx))

    @staticmethod
    def _apply_pair_op(op, lv, rv, valid_l, valid_r):
        if op == "sum":
            return lv + rv
        if op == "left":
            return lv
        if op == "right":
            return rv
        if op == "mul":
            return lv * rv
        raise ValueError(
            "join aggregate not columnar ({!r}); use funcs-recognized "
            "aggregates or the host engine".format(op))

    # -- sink --------------------------------------------------------------

    def run_sink(self, stage, ins):
        """Sink semantics match the host SinkWriter: one text line per
        record, value only, any value str-formatted (reference:
        dataset.py:277-278 prints the value).  Untagged sinks (pending
        format maps, e.g. sink_tsv's) run the mapper on host over the
        decoded records.  Returns a lazy view over the written part file
        so the sink's output is readable downstream, matching
        _SinkWorker (runner.py:278-298)."""
        import os
        path = stage.path
        os.makedirs(path, exist_ok=True)
        ds = self._collect(self._merge_stores(ins))
        fname = os.path.join(path, "part-{}".format(self.rank))
        if stage.options.get("device_sink") == "values":
            with open(fname, "w") as fh:
                for _kk, vv in ds.read():
                    fh.write("{}\n".format(vv))
        else:
            from ..dataset import MemoryDataset
            mem = MemoryDataset(list(ds.read()))
            with open(fname, "w") as fh:
                for _kk, vv in stage.mapper.map(mem):
                    fh.write("{}\n".format(vv))
        return SinkStore([fname])

    # -- host fallback -----------------------------------------------------

    def _decode_store(self, store):
        if isinstance(store, HostStore):
            return list(store)
        if isinstance(store, SinkStore):
            return [kv for ds in store.datasets() for kv in ds.read()]
        tbl = getattr(store, "str_table", None)
        if isinstance(store, TextSource):
            records = []
            pos = 0
            data = store.text.tobytes()
            for line in data.split(b"\n"):
                if line or pos + len(line) < len(data):
                    records.append(
                        (pos, line.decode("utf-8", "replace")))
                pos += len(line) + 1
            if records and pos - 1 >= len(data) and not records[-1][1]:
                records.pop()
            return records
        if isinstance(store, TokenStore):
            return list(TokenColumnDataset(
                store, getattr(store, "keyed", False)).read())
        records = []
        keyed = getattr(store, "keyed", False)
        fkeys = getattr(store, "fkeys", False)
        for p in sorted(store):
            keys, vals = self._merged_partition([store], p)
            if keys is None:
                continue
            if fkeys:
                keys = _decode_f64_sortable(keys)
            kl = keys.cpu().tolist()
            if tbl is not None:
                kl = [tbl[k] for k in kl]
            vl = vals.cpu().tolist()
            if keyed:
                records.extend((k, (k, v)) for k, v in zip(kl, vl))
            else:
                records.extend(zip(kl, vl))
        return records

    def _host_map(self, stage, ins):
        """Decode -> run the stage's (opaque) mapper on host -> re-encode.
        The shuffle core stays on device either side of this boundary.
        Input 0 is the primary dataset; the rest ride as supplemental
        dataset-lists (cross joins — reference: stagerunner.py:66-74).
        Concat stages treat every input as a primary."""
        from ..dataset import MemoryDataset
        if stage.options.get("concat"):
            out_records = []
            for store in ins:
                out_records.extend(stage.mapper.map(
                    MemoryDataset(self._decode_store(store))))
            return self._encode_or_host(out_records)
        primary = MemoryDataset(self._decode_store(ins[0]))
        supplemental = []
        for s in ins[1:]:
            recs = self._decode_store(s)
            if self.world > 1:
                # supplemental sides (cross/broadcast joins) must be
                # COMPLETE on every rank: computed stores are rank-owned
                # shards and raw inputs are rank slices, so gather the
                # union (the RCCL-bcast analog of K9's broadcast side)
                recs = self._host_gather(recs)
            supplemental.append([MemoryDataset(recs)])
        out_records = list(stage.mapper.map(primary, *supplemental))
        return self._encode_or_host(out_records)

    def _host_gather(self, recs):
        """Union of every rank's records, identical on all ranks (one
        collective; deterministic rank order)."""
        import torch.distributed as dist
        gathered = [None] * self.world
        dist.all_gather_object(gathered, recs)
        return [r for lst in gathered for r in lst]

    def _host_reduce(self, stage, ins):
        from ..dataset import MemoryDataset
        datasets = []
        for store in ins:
            recs = self._decode_store(store)
            if self.world > 1:
                recs = self._host_exchange(recs)
            recs.sort(key=lambda r: r[0])     # reducers need sorted streams
            datasets.append([MemoryDataset(recs)])
        out_records = list(stage.reducer.reduce(*datasets))
        return self._encode_or_host(out_records)

    def _host_exchange(self, recs):
        """Host-record analog of the column exchange (one collective per
        input store, deterministic across ranks): gather every rank's
        records, keep the keys this rank owns by the process-stable
        hash.  Without this, a multi-rank host-fallback reduce would
        silently fold only the local slice of each key's values."""
        import torch.distributed as dist
        from ..keyhash import stable_hash64
        gathered = [None] * self.world
        dist.all_gather_object(gathered, recs)
        return [r for lst in gathered for r in lst
                if stable_hash64(r[0]) % self.world == self.rank]
