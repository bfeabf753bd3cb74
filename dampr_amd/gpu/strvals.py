"""Variable-length (string) VALUE columns for the device engine.

A ``StrVals`` column is a byte arena: one contiguous u8 ``blob`` plus an
``offs`` int64 prefix array (n+1 entries).  This is the SURVEY §7
"variable-length values" design: values move through partition, sort,
spill, exchange and join as opaque bytes addressed by row index — only
keys are ever compared on device (reference analog: the opaque pickled
value streams of dampr/dataset.py:119-159, re-laid-out for HBM).

All ops are torch-tensor ops (work on CPU for tests and CUDA for the
engine); the row gather — the hot reorder under sort/join/partition —
uses the wave-per-row HIP kernel when the extension is loaded on a CUDA
device, with a pure-torch fallback as the CPU oracle.  (Reference
analog: values are opaque pickled streams the engine never interprets,
dampr/dataset.py:119-159; here they are opaque BYTES the kernels never
interpret — only keys compare on device.)
"""
import numpy as np
import torch


class StrVals(object):
    """Immutable var-len value column: blob (u8) + offs (i64, n+1)."""

    __slots__ = ("blob", "offs")

    is_strvals = True

    def __init__(self, blob, offs):
        assert blob.dtype == torch.uint8
        assert offs.dtype == torch.int64
        self.blob = blob
        self.offs = offs

    # ---- constructors ----------------------------------------------------

    @classmethod
    def from_strings(cls, strings, device=None):
        """Encode a list/array of Python strings (UTF-8)."""
        bs = [s.encode("utf-8") for s in strings]
        lens = np.fromiter((len(b) for b in bs), dtype=np.int64,
                           count=len(bs))
        offs = np.zeros(len(bs) + 1, dtype=np.int64)
        np.cumsum(lens, out=offs[1:])
        blob = np.frombuffer(b"".join(bs), dtype=np.uint8).copy() \
            if bs else np.zeros(0, dtype=np.uint8)
        bt = torch.from_numpy(blob)
        ot = torch.from_numpy(offs)
        if device is not None:
            bt, ot = bt.to(device), ot.to(device)
        return cls(bt, ot)

    @classmethod
    def empty(cls, device=None):
        return cls(torch.zeros(0, dtype=torch.uint8, device=device),
                   torch.zeros(1, dtype=torch.int64, device=device))

    # ---- tensor-like surface the engine relies on ------------------------

    @property
    def device(self):
        return self.blob.device

    @property
    def dtype(self):
        return torch.uint8          # sentinel; engine checks is_strvals

    def numel(self):
        return self.offs.numel() - 1

    def element_size(self):
        return 1

    @property
    def nbytes(self):
        return self.blob.numel() + self.offs.numel() * 8

    def __len__(self):
        return self.numel()

    def __getitem__(self, idx):
        """Row gather: StrVals[idx_tensor] -> StrVals (the reorder op
        under sort permutations, join row indices and partition order).
        Slices take the fast contiguous path."""
        if isinstance(idx, slice):
            lo, hi, step = idx.indices(self.numel())
            assert step == 1
            b0 = self.offs[lo]
            sub_off = self.offs[lo:hi + 1] - b0
            blob = self.blob[int(b0.item()):int(self.offs[hi].item())]
            return StrVals(blob.contiguous(), sub_off.contiguous())
        return self.gather(idx)

    def to(self, device, non_blocking=False):
        return StrVals(self.blob.to(device, non_blocking=non_blocking),
                       self.offs.to(device, non_blocking=non_blocking))

    def clone(self):
        return StrVals(self.blob.clone(), self.offs.clone())

    def contiguous(self):
        return self

    # ---- core ops --------------------------------------------------------

    def lens(self):
        return self.offs[1:] - self.offs[:-1]

    def gather(self, idx):
        """New StrVals with rows reordered/selected by ``idx`` (i64)."""
        idx = idx.to(torch.int64)
        n = idx.numel()
        if n == 0:
            return StrVals.empty(self.blob.device)
        src_off = self.offs[idx]
        ln = self.offs[idx + 1] - src_off
        new_offs = torch.zeros(n + 1, dtype=torch.int64,
                               device=idx.device)
        torch.cumsum(ln, 0, out=new_offs[1:])
        total = int(new_offs[-1].item())
        if total == 0:
            return StrVals(
                torch.zeros(0, dtype=torch.uint8, device=idx.device),
                new_offs)
        if self.blob.is_cuda:
            from ..ops import native
            ext = native.require()
            out = torch.empty(total, dtype=torch.uint8,
                              device=self.blob.device)
            ext.varlen_gather(self.blob, src_off, ln, new_offs, out)
            return StrVals(out, new_offs)
        # torch oracle: per-output-byte source index
        row_of_byte = torch.repeat_interleave(
            torch.arange(n, device=idx.device), ln)
        within = torch.arange(total, device=idx.device) \
            - new_offs[row_of_byte]
        src = src_off[row_of_byte] + within
        return StrVals(self.blob[src], new_offs)

    @staticmethod
    def cat(cols):
        cols = list(cols)
        if not cols:
            return StrVals.empty()
        if len(cols) == 1:
            return cols[0]
        dev = cols[0].blob.device
        blob = torch.cat([c.blob for c in cols])
        offs = [cols[0].offs]
        base = cols[0].offs[-1]
        for c in cols[1:]:
            offs.append(c.offs[1:] + base)
            base = base + c.offs[-1]
        return StrVals(blob, torch.cat(offs).to(dev))

    # ---- host conversion -------------------------------------------------

    def tolist(self):
        """Decode to Python strings."""
        b = self.blob.cpu().numpy().tobytes()
        offs = self.offs.cpu().numpy()
        return [b[offs[i]:offs[i + 1]].decode("utf-8", "replace")
                for i in range(len(offs) - 1)]

    def cpu(self):
        return self.to("cpu")

    def record_stream(self, stream):
        """Cross-stream lifetime marker (HbmPool.prefetch)."""
        self.blob.record_stream(stream)
        self.offs.record_stream(stream)
