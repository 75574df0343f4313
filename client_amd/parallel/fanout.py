"""Multi-replica fan-out over RCCL/xGMI.

The BASELINE 8-GPU configs stage one input on a root rank and replicate
it to every GPU replica's HIP-shm region before the per-replica infer
requests go out (xGMI is point-to-point, 7 links per
GPU — RCCL picks tree/direct algorithms for the one-to-eight
broadcast). ``torch.distributed`` with the nccl backend IS RCCL on
ROCm; on CPU test boxes the same code runs over gloo.
"""

import os

import torch
import torch.distributed as dist


def init_distributed(backend=None):
    """Initialize torch.distributed from torchrun env vars; no-op and
    returns (rank 0, world 1) when not launched distributed."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend)
    return rank, world


class RegionBroadcaster:
    """Broadcast one staged tensor (usually a HIP-shm region) from the
    root rank to every replica's copy.

    Accepts either a torch tensor or a hip_shared_memory region handle
    (wrapped zero-copy via DLPack, kDLROCM) — the collective operates
    directly on HBM-resident region memory; no staging copies.
    """

    def __init__(self, target, shape=None, datatype="BF16", src=0):
        self.src = src
        if isinstance(target, torch.Tensor):
            self._tensor = target
        else:
            import client_amd.utils.hip_shared_memory as hipshm

            smt = hipshm.as_shared_memory_tensor(target, datatype, list(shape))
            self._tensor = torch.from_dlpack(smt)

    @property
    def tensor(self):
        return self._tensor

    def broadcast(self, async_op=False):
        """Fan the root's staged data out to all replicas."""
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return None
        return dist.broadcast(self._tensor, src=self.src, async_op=async_op)


def aggregate_max(value, device=None):
    """Max of a scalar across ranks (whole-job elapsed time uses the
    slowest rank). Returns value unchanged when not distributed."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return value
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    t = torch.tensor([float(value)], device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


class OverlappedBroadcaster:
    """Ping-pong fan-out pipeline: pack(n+1) into buffer B overlaps the
    RCCL broadcast of buffer A on a side HIP stream, event-ordered, and
    request issue is host-gated on the broadcast-done event ONLY — no
    per-step global synchronize (the design BASELINE.json names; the
    host gate is required because the server is a separate process whose
    streams cannot wait on this process's events).

    Per step n:
      wait_ready()                       # hipEvent host-wait: bcast(n-1)
      stage_and_broadcast(n % 2, pack)   # pack on stream0, bcast on the
                                         # side stream after a pack event
      ... issue step-n requests against buffer (n-1) % 2 ...

    ``targets`` are the ping-pong staging buffers: torch tensors or
    hip_shared_memory region handles (wrapped zero-copy via DLPack so
    the collective reads/writes HBM-resident region memory directly).

    Broadcast times come from hipEvents bracketing the collective on the
    side stream (``bcast_ms``). On CPU (gloo tests) the same control
    flow runs with synchronous broadcasts and no events.
    """

    def __init__(self, targets, shape=None, datatype="BF16", src=0,
                 pack_stream_handle=None):
        self.src = src
        self._tensors = []
        for target in targets:
            if isinstance(target, torch.Tensor):
                self._tensors.append(target)
            else:
                import client_amd.utils.hip_shared_memory as hipshm

                smt = hipshm.as_shared_memory_tensor(
                    target, datatype, list(shape)
                )
                self._tensors.append(torch.from_dlpack(smt))
        self._cuda = self._tensors[0].is_cuda
        self.bcast_ms = []
        self._inflight = False
        if self._cuda:
            if pack_stream_handle is None:
                from ..ops import hip_runtime as hr

                pack_stream_handle = hr.stream_handle(
                    self._tensors[0].device.index or 0, 0
                )
            self._pack_stream = torch.cuda.ExternalStream(
                int(pack_stream_handle)
            )
            self._side = torch.cuda.Stream()
            self._pack_done = torch.cuda.Event()
            self._ev_start = torch.cuda.Event(enable_timing=True)
            self._ev_end = torch.cuda.Event(enable_timing=True)

    def tensor(self, idx=0):
        return self._tensors[idx]

    @property
    def _active(self):
        # any initialized group, INCLUDING world=1: a single-rank
        # broadcast is a valid (cheap) collective, and running it keeps
        # the whole event/side-stream machinery exercised by the
        # world=1 GPU test instead of silently skipped (bench.py only
        # constructs a broadcaster when world > 1, so the N=1 headline
        # path never pays for this)
        return dist.is_initialized()

    def stage_and_broadcast(self, buf_idx, pack_fn=None):
        """Enqueue pack (caller's fn, async on the pack stream) and the
        broadcast of buffer ``buf_idx`` (side stream, ordered after the
        pack by event). Returns immediately — nothing here blocks the
        host or the serving streams."""
        if pack_fn is not None:
            pack_fn(buf_idx)
        if not self._active:
            return
        t = self._tensors[buf_idx]
        if self._cuda:
            self._pack_done.record(self._pack_stream)
            self._side.wait_event(self._pack_done)
            with torch.cuda.stream(self._side):
                self._ev_start.record()
                work = dist.broadcast(t, src=self.src, async_op=True)
                work.wait()  # side stream waits on the RCCL stream
                self._ev_end.record()
            self._inflight = True
        else:
            dist.broadcast(t, src=self.src)

    def wait_ready(self):
        """Host-gate on the last broadcast's completion (only the side
        stream — not the device); records the hipEvent-measured bcast
        duration. Uses hipStreamSynchronize rather than
        hipEventSynchronize: host event-waits measured pathologically
        slow on this ROCm stack (profiles/sync_mode_ab_r02.md)."""
        if self._inflight:
            self._side.synchronize()
            self.bcast_ms.append(self._ev_start.elapsed_time(self._ev_end))
            self._inflight = False


class PeerScatterBroadcaster:
    """8-way fan-out alternative: the root copies its staged buffer to
    every peer's HIP-IPC region with one hipMemcpyPeerAsync per
    destination, each on its own stream — the 7 copies ride 7 distinct
    xGMI links concurrently instead of being per-link bound like a ring
    (SURVEY.md §2.8). Peers learn completion through a gloo-side
    barrier (host signal; their server processes read the regions
    independently, so the gate must be host-side anyway).

    Selected in bench.py with CLIENT_AMD_FANOUT=p2p; requires every
    rank's region to be IPC-shareable. Same stage/wait interface as
    OverlappedBroadcaster so the two are interchangeable for
    measurement.
    """

    SCATTER_STREAM0 = 2  # hip_runtime cached-stream indices 2..2+N

    def __init__(self, region_handles, nbytes, src=0):
        from ..ops import hip_runtime as hr

        self._hr = hr
        self.src = src
        self.nbytes = nbytes
        self.bcast_ms = []
        self._inflight = False
        self._t0 = 0.0
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        world = dist.get_world_size() if dist.is_initialized() else 1
        # host-side completion signaling (cheap; no GPU involvement)
        self._signal_group = (
            dist.new_group(backend="gloo") if world > 1 else None
        )
        self._local = region_handles[0] if region_handles else None
        self._peers = []
        if world > 1:
            # exchange (ipc handle, device, ptr) so the root can open
            # every peer's region
            import client_amd.utils.hip_shared_memory as hipshm

            mine = [
                (hipshm.get_raw_handle_bytes(h), h._device_id)
                for h in region_handles
            ]
            allh = [None] * world
            dist.all_gather_object(allh, mine)
            if self.rank == src:
                for r, handles in enumerate(allh):
                    if r == src:
                        continue
                    for hnd, dev in handles:
                        ptr = hr.ipc_open_mem_handle(hnd)
                        self._peers.append((ptr, dev))

    def stage_and_broadcast(self, buf_idx=0, pack_fn=None):
        import time as _time

        if pack_fn is not None:
            pack_fn(buf_idx)
        if self._signal_group is None:
            return
        if self.rank == self.src and self._local is not None:
            hr = self._hr
            dev = self._local._device_id
            n_streams = len(self._peers)
            idxs = [self.SCATTER_STREAM0 + i for i in range(n_streams)]
            # order every scatter stream after the pack (stream 0)
            hr.stream_fence(dev, 0, idxs)
            src_ptr = self._local._base_addr
            for i, (ptr, peer_dev) in enumerate(self._peers):
                hr.memcpy_peer_async(ptr, peer_dev, src_ptr, dev,
                                     self.nbytes, idxs[i])
        self._t0 = _time.monotonic()
        self._inflight = True

    def wait_ready(self):
        import time as _time

        if not self._inflight:
            return
        if self.rank == self.src and self._local is not None:
            hr = self._hr
            dev = self._local._device_id
            for i in range(len(self._peers)):
                hr.stream_sync(dev, self.SCATTER_STREAM0 + i)
        if self._signal_group is not None:
            dist.barrier(group=self._signal_group)
        self.bcast_ms.append((_time.monotonic() - self._t0) * 1e3)
        self._inflight = False
