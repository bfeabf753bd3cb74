"""Multi-process (gloo, CPU) tests of the distributed exchange logic — the
same code path the RCCL/xGMI shuffle uses, minus the transport."""
import multiprocessing
import os

import numpy as np
import pytest
import torch


def _run_rank(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from dampr_amd.parallel.shuffle import (exchange_pairs,
                                                exchange_keyed_payload,
                                                all_reduce_scalar, owner_of)

        g = torch.Generator().manual_seed(100 + rank)
        keys = torch.randint(1, 1000, (500,), dtype=torch.int64,
                             generator=g)
        vals = torch.ones(500, dtype=torch.int64)
        rk, rv = exchange_pairs(keys, vals)
        # every received key belongs to us
        assert bool((owner_of(rk, world) == rank).all())

        # payload exchange: token bytes are the key repeated (key % 5 + 1)x
        lens = (keys % 5 + 1).to(torch.int64)
        blob = torch.repeat_interleave((keys % 251).to(torch.uint8), lens)
        ek, ev, eb, el = exchange_keyed_payload(keys, vals, blob, lens)
        assert bool((owner_of(ek, world) == rank).all())
        assert int(el.sum().item()) == eb.numel()
        # payload integrity: each slice is its key's byte repeated
        offs = torch.cumsum(el, 0) - el
        for i in range(min(50, ek.numel())):
            o, ln = int(offs[i]), int(el[i])
            want = int(ek[i] % 251)
            assert eb[o:o + ln].eq(want).all()

        total = all_reduce_scalar(int(vals.sum().item()))
        assert total == world * 500

        # global conservation of (key, count) mass
        local = torch.zeros(1000, dtype=torch.int64)
        local.index_add_(0, rk, rv)
        dist.all_reduce(local)
        want = torch.zeros(1000, dtype=torch.int64)
        want_local = torch.zeros(1000, dtype=torch.int64)
        want_local.index_add_(0, keys, vals)
        dist.all_reduce(want_local)
        assert torch.equal(local, want_local)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:      # noqa: BLE001
        import traceback
        q.put((rank, traceback.format_exc()))


@pytest.mark.parametrize("world", [2, 3])
def test_exchange_gloo(world):
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + (os.getpid() + 90 + world) % 900
    procs = [ctx.Process(target=_run_rank, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", "rank {} failed:\n{}".format(rank, status)


def _rank_varlen(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from dampr_amd.gpu.strvals import StrVals
        from dampr_amd.parallel.shuffle import exchange_columns_varlen
        rng = torch.Generator().manual_seed(100 + rank)
        n = 500 + rank * 7
        keys = torch.randint(0, 1000, (n,), generator=rng)
        pids = torch.randint(0, 64, (n,), generator=rng)
        # value i is "<key>:<pid>" so integrity is checkable post-route
        sv = StrVals.from_strings(
            ["{}:{}".format(int(k), int(p))
             for k, p in zip(keys, pids)])
        rk, rsv, rp = exchange_columns_varlen(keys, sv, pids, world)
        assert bool((torch.remainder(rp, world) == rank).all())
        vals = rsv.tolist()
        assert len(vals) == rk.numel()
        for i in range(rk.numel()):
            assert vals[i] == "{}:{}".format(int(rk[i]), int(rp[i]))
        # conservation: total rows across ranks
        tot = torch.tensor([rk.numel()])
        dist.all_reduce(tot)
        want = torch.tensor([sum(500 + r * 7 for r in range(world))])
        assert torch.equal(tot, want)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:       # noqa: BLE001
        import traceback
        q.put((rank, traceback.format_exc()))


@pytest.mark.parametrize("world", [2, 3])
def test_exchange_varlen_gloo(world):
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + (os.getpid() + 170 + world) % 900
    procs = [ctx.Process(target=_rank_varlen, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", "rank {} failed:\n{}".format(rank, status)
