"""client_amd.parallel — multi-replica fan-out over RCCL/xGMI."""

from .fanout import (
    OverlappedBroadcaster,
    PeerScatterBroadcaster,
    RegionBroadcaster,
    aggregate_max,
    init_distributed,
)

__all__ = [
    "OverlappedBroadcaster",
    "PeerScatterBroadcaster",
    "RegionBroadcaster",
    "aggregate_max",
    "init_distributed",
]
