"""Per-stage tracing: structured timings for every stage a runner
executes, with HIP-event device timing on the GPU engine.

The reference has no tracing at all (SURVEY.md §5.1 — logging only); this
is the promised upgrade.  Usage:

    from dampr_amd.utils.trace import get_trace
    emitter = pipeline.run()
    for row in get_trace().rows:      # most recent run
        print(row)

Runners call ``trace_stage`` around each stage; on a CUDA device the
context also brackets the stage with HIP events so the device time
(kernel + copy work enqueued by the stage) is reported separately from
wall time.
"""
import contextlib
import logging
import time

log = logging.getLogger("dampr_amd")


class Trace(object):
    def __init__(self):
        self.rows = []

    def clear(self):
        self.rows = []

    def add(self, row):
        self.rows.append(row)

    def report(self):
        lines = ["{:<40} {:>10} {:>10}".format("stage", "wall_ms",
                                               "device_ms")]
        for r in self.rows:
            dev = ("{:.2f}".format(r["device_ms"])
                   if r.get("device_ms") is not None else "-")
            lines.append("{:<40} {:>10.2f} {:>10}".format(
                r["stage"][:40], r["wall_ms"], dev))
        return "\n".join(lines)


_TRACE = Trace()


def get_trace():
    return _TRACE


@contextlib.contextmanager
def trace_stage(name, device=None):
    """Time a stage; on a CUDA device also record HIP-event device time."""
    ev_start = ev_end = None
    if device is not None and getattr(device, "type", None) == "cuda":
        import torch
        ev_start = torch.cuda.Event(enable_timing=True)
        ev_end = torch.cuda.Event(enable_timing=True)
        ev_start.record()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        wall_ms = (time.perf_counter() - t0) * 1000.0
        device_ms = None
        if ev_start is not None:
            import torch
            ev_end.record()
            ev_end.synchronize()
            device_ms = ev_start.elapsed_time(ev_end)
        row = {"stage": str(name), "wall_ms": wall_ms,
               "device_ms": device_ms}
        _TRACE.add(row)
        log.debug("stage %s: %.2f ms wall%s", name, wall_ms,
                  " / %.2f ms device" % device_ms
                  if device_ms is not None else "")
