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
