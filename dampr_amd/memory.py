"""Adaptive RSS watermark governor for spill decisions.

Role parity with the reference's ``dampr/memory.py`` (high-water RSS check
with amortized frequency; reference: memory.py:12-122).  This implementation
is a single adaptive controller: after every real RSS reading it estimates
bytes/record from the delta and schedules the next check at half the
projected distance to the watermark, clamped to the settings bounds.  The
reference's broken ``ExponentialMemoryChecker`` has no
analog here.

The GPU engine applies the same watermark idea over HBM pool occupancy
instead of RSS (gpu/engine.py ``HbmPool``).
"""
import os

from . import settings

_PAGE = os.sysconf("SC_PAGE_SIZE") if hasattr(os, "sysconf") else 4096


def rss_mb() -> float:
    """Current resident set size of this process, in MB."""
    try:
        with open("/proc/self/statm", "rb") as fh:
            return int(fh.read().split()[1]) * _PAGE / (1024.0 * 1024.0)
    except OSError:
        return 0.0


class MemoryGovernor(object):
    """Answers "should this buffer spill now?" cheaply per record."""

    def __init__(self, limit_mb=None):
        self.limit_mb = float(limit_mb if limit_mb is not None
                              else settings.max_memory_per_worker)
        self.base_mb = rss_mb()
        self._count = 0
        self._next_check = min(settings.memory_min_count,
                               settings.memory_max_count_before_check)
        self._last_count = 0
        self._last_mb = self.base_mb

    def over_watermark_bulk(self, n) -> bool:
        """Amortized form: account n records at once."""
        self._count += n - 1
        return self.over_watermark()

    def over_watermark(self) -> bool:
        """Call once per record added; True when the buffer should spill."""
        self._count += 1
        if self._count < self._next_check:
            return False
        cur = rss_mb()
        used = cur - self.base_mb
        if used >= self.limit_mb:
            return True
        # Estimate records until the watermark and check halfway there.
        d_records = max(1, self._count - self._last_count)
        d_mb = max(0.0, cur - self._last_mb)
        mb_per_record = d_mb / d_records
        if mb_per_record > 0:
            remaining = (self.limit_mb - used) / mb_per_record
            step = int(remaining / 2)
        else:
            step = settings.memory_max_count_before_check
        step = max(min(step, settings.memory_max_count_before_check),
                   settings.memory_min_count)
        self._last_count = self._count
        self._last_mb = cur
        self._next_check = self._count + step
        return False

    def reset(self):
        """After a spill: re-baseline so freed memory is accounted for."""
        self._count = 0
        self._last_count = 0
        self._last_mb = rss_mb()
        self._next_check = min(settings.memory_min_count,
                               settings.memory_max_count_before_check)
