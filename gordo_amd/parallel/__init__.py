from .fleet import build_fleet, shard_machines, init_distributed
from .packed_builder import PackedFleetBuilder

__all__ = [
    "build_fleet",
    "shard_machines",
    "init_distributed",
    "PackedFleetBuilder",
]
