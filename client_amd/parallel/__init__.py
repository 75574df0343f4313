"""client_amd.parallel — multi-replica fan-out over RCCL/xGMI."""

from .fanout import RegionBroadcaster, aggregate_max, init_distributed

__all__ = ["RegionBroadcaster", "aggregate_max", "init_distributed"]
