"""Data-store types (reference parity: data_store/types.py)."""
from dataclasses import dataclass, field


@dataclass
class BroadcastWindow:
    """Coordinates a W-party tensor/file broadcast: participants join the
    group; when world_size is reached the transfer manifests are executed."""
    world_size: int = 2
    timeout: float = 300.0
    group_id: str = None
    ips: list = field(default_factory=list)
    fanout: int = 2        # 2 for GPU (RCCL tree), ~50 for fs broadcast
    pack: bool = False     # pack state dicts into one flat buffer


class Locale:
    STORE = "store"   # data lives on the central store pod
    LOCAL = "local"   # zero-copy: registered in metadata, served p2p


class Lifespan:
    CLUSTER = "cluster"    # persists on the store PVC
    RESOURCE = "resource"  # lives with the owning workload
