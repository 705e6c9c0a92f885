from .engine import (
    DistributedEngine,
    GalleryResult,
    GenerationRequest,
    LocalEngine,
)
from .group import (
    allgather_floats,
    barrier,
    broadcast_object,
    destroy_group,
    gather_images,
    init_group,
    sync_weights,
)

__all__ = [
    "DistributedEngine",
    "GalleryResult",
    "GenerationRequest",
    "LocalEngine",
    "allgather_floats",
    "barrier",
    "broadcast_object",
    "destroy_group",
    "gather_images",
    "init_group",
    "sync_weights",
]
