"""Multi-process fan-out tests (gloo, world_size=2, CPU) — the
distributed control flow bench.py runs over RCCL on the 8-GPU node."""

import multiprocessing as mp
import os

import numpy as np
import pytest

torch = pytest.importorskip("torch")


def _rank_main(rank, world, port, conn):
    try:
        os.environ.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
        })
        import torch
        import torch.distributed as dist

        from client_amd.parallel import (
            RegionBroadcaster,
            aggregate_max,
            init_distributed,
        )

        r, w = init_distributed(backend="gloo")
        assert (r, w) == (rank, world)
        # each replica stages its own buffer; root's content wins
        staged = torch.full((64,), float(rank + 1))
        bc = RegionBroadcaster(staged, src=0)
        bc.broadcast()
        assert torch.all(bc.tensor == 1.0), "broadcast did not replicate root"

        slowest = aggregate_max(10.0 + rank, device="cpu")
        assert slowest == 10.0 + world - 1

        dist.destroy_process_group()
        conn.send("ok")
    except Exception as e:  # pragma: no cover
        conn.send(f"error rank {rank}: {e}")


def test_region_broadcast_gloo_world2():
    ctx = mp.get_context("spawn")
    port = 29712
    procs = []
    conns = []
    for rank in range(2):
        parent, child = ctx.Pipe()
        p = ctx.Process(target=_rank_main, args=(rank, 2, port, child))
        p.start()
        procs.append(p)
        conns.append(parent)
    for rank, (p, conn) in enumerate(zip(procs, conns)):
        assert conn.poll(120), f"rank {rank} timed out"
        msg = conn.recv()
        p.join(30)
        assert msg == "ok", msg


def test_broadcaster_single_process_noop():
    from client_amd.parallel import RegionBroadcaster, aggregate_max

    t = torch.ones(8)
    bc = RegionBroadcaster(t)
    assert bc.broadcast() is None  # no dist group -> no-op
    assert aggregate_max(3.5, device="cpu") == 3.5


def test_region_broadcast_gloo_world4():
    """World=4 replica fan-out (the 8-GPU shape at CI scale): broadcast +
    max-aggregation across two ranks per... four single-slot replicas."""
    ctx = mp.get_context("spawn")
    port = 29716
    procs = []
    conns = []
    for rank in range(4):
        parent, child = ctx.Pipe()
        p = ctx.Process(target=_rank_main, args=(rank, 4, port, child))
        p.start()
        procs.append(p)
        conns.append(parent)
    for rank, (p, conn) in enumerate(zip(procs, conns)):
        assert conn.poll(180), f"rank {rank} timed out"
        msg = conn.recv()
        assert msg == "ok", msg
    for p in procs:
        p.join(timeout=30)
