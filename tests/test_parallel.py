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


def _overlap_rank_main(rank, world, port, conn):
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
            OverlappedBroadcaster,
            init_distributed,
        )

        init_distributed(backend="gloo")
        # ping-pong staging buffers, exactly the bench.py pipeline:
        # prologue stages buffer 0; step n gates on bcast(n), stages
        # (n+1)%2, then "serves" from buffer n%2
        buffers = [torch.zeros(32), torch.zeros(32)]
        bc = OverlappedBroadcaster(buffers, src=0)

        def pack_fn(buf_idx):
            # only the root's pack content matters (value = step number
            # it is staged for, recoverable at serve time)
            buffers[buf_idx].fill_(float(rank * 1000 + pack_fn.step))

        pack_fn.step = 0
        bc.stage_and_broadcast(0, pack_fn)
        served = []
        for n in range(6):
            bc.wait_ready()
            pack_fn.step = n + 1
            bc.stage_and_broadcast((n + 1) % 2, pack_fn)
            # serve step n from buffer n%2: root packed value n for it
            served.append(float(buffers[n % 2][0]))
        assert served == [float(i) for i in range(6)], served

        dist.destroy_process_group()
        conn.send("ok")
    except Exception as e:  # pragma: no cover
        conn.send(f"error rank {rank}: {e}")


def test_overlapped_broadcaster_pingpong_gloo_world2():
    """The bench.py world>1 pipeline at CI scale: every rank must see
    the root's step-n staging in buffer n%2 when it serves step n, with
    the next stage+broadcast already enqueued."""
    ctx = mp.get_context("spawn")
    port = 29718
    procs = []
    conns = []
    for rank in range(2):
        parent, child = ctx.Pipe()
        p = ctx.Process(target=_overlap_rank_main,
                        args=(rank, 2, port, child))
        p.start()
        procs.append(p)
        conns.append(parent)
    for rank, (p, conn) in enumerate(zip(procs, conns)):
        assert conn.poll(120), f"rank {rank} timed out"
        msg = conn.recv()
        p.join(30)
        assert msg == "ok", msg


def test_overlapped_broadcaster_single_process():
    from client_amd.parallel import OverlappedBroadcaster

    buffers = [torch.zeros(8), torch.zeros(8)]
    bc = OverlappedBroadcaster(buffers)
    bc.stage_and_broadcast(0, lambda i: buffers[i].fill_(7.0))
    bc.wait_ready()  # no dist group: both are no-ops beyond pack
    assert torch.all(buffers[0] == 7.0)
    assert bc.bcast_ms == []


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
