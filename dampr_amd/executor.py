"""Per-stage worker pools (the CPU "communication backend").

Role parity with the reference's ``dampr/stagerunner.py`` fork+Queue pools,
with two deliberate upgrades over the reference:

* **Fault detection**: a crashed worker deadlocks the reference forever
  (reference: stagerunner.py:35-37).  Here the parent polls
  child liveness while collecting results and raises with the child's
  traceback instead of hanging.
* Workers are small classes with a setup/process/finish lifecycle instead of
  four near-duplicate ``*StageRunner`` subclasses.

Processes are forked so closures (user lambdas) never cross a pickle
boundary; only job descriptors and {partition: [dataset-handle]} results do.
"""
import logging
import multiprocessing
import queue as queue_mod
import traceback

log = logging.getLogger("dampr_amd")

_ctx = multiprocessing.get_context("fork")


class StageWorker(object):
    """One worker's life: setup(worker_fs) once, process(job) per job,
    finish() -> result sent back to the parent."""

    def setup(self, worker_id, stage_fs):
        pass

    def process(self, job):
        raise NotImplementedError()

    def finish(self):
        raise NotImplementedError()


def _child_main(worker, w_id, stage_fs, in_q, out_q):
    try:
        worker.setup(w_id, stage_fs)
        while True:
            job = in_q.get()
            if job is None:
                break
            worker.process(job)
        out_q.put(("ok", worker.finish()))
    except BaseException:
        try:
            out_q.put(("err", traceback.format_exc()))
        except Exception:
            pass
        raise


class WorkerCrash(RuntimeError):
    pass


def run_stage(worker_factory, jobs, n_procs, stage_fs):
    """Run ``jobs`` across ``n_procs`` forked workers; returns the list of
    worker finish() results."""
    from .base import Splitter
    Splitter.check_start_method()   # fork-only invariant, fail loudly
    in_q = _ctx.Queue()
    out_q = _ctx.Queue()
    n_jobs = 0
    for job in jobs:
        in_q.put(job)
        n_jobs += 1
    n_procs = max(1, min(n_procs, max(n_jobs, 1)))
    log.debug("stage: %d jobs across %d workers", n_jobs, n_procs)

    procs = []
    for w_id in range(n_procs):
        worker = worker_factory()
        p = _ctx.Process(target=_child_main,
                         args=(worker, w_id, stage_fs, in_q, out_q))
        p.daemon = True
        p.start()
        in_q.put(None)          # one sentinel per worker
        procs.append(p)

    results = []
    errors = []
    pending = set(range(n_procs))
    while pending:
        try:
            status, payload = out_q.get(timeout=1.0)
        except queue_mod.Empty:
            dead = [p for p in procs if not p.is_alive()]
            if len(dead) > len(results) + len(errors) and out_q.empty():
                for p in procs:
                    p.terminate()
                raise WorkerCrash(
                    "worker process died (exit codes: {})".format(
                        [p.exitcode for p in dead]))
            continue
        pending.pop()
        if status == "ok":
            results.append(payload)
        else:
            errors.append(payload)
    for p in procs:
        p.join()
    if errors:
        raise WorkerCrash("worker failed:\n" + "\n".join(errors))
    return results


def run_stage_inline(worker_factory, jobs, stage_fs):
    """Single-process fallback (debugging, n_procs=0)."""
    worker = worker_factory()
    worker.setup(0, stage_fs)
    for job in jobs:
        worker.process(job)
    return [worker.finish()]
