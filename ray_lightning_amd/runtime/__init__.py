"""Native actor runtime: process actors, object store, queue, resources.

This package owns the substrate the reference delegated to Ray core
(actors, object store, ``ray.util.queue``, the resource scheduler)."""
from .actor import (ActorFuture, ActorHandle, RemoteError, get_gpu_ids,
                    get_node_and_gpu_ids, get_node_ip)
from .object_store import ObjectRef, ObjectStore
from .queue import Queue
from .resources import (GpuAllocator, PlacementGroup, ResourceError,
                        node_cpu_count, visible_gpu_ids)

__all__ = [
    "ActorFuture", "ActorHandle", "RemoteError", "get_gpu_ids",
    "get_node_and_gpu_ids", "get_node_ip", "ObjectRef", "ObjectStore",
    "Queue", "GpuAllocator", "PlacementGroup", "ResourceError",
    "node_cpu_count", "visible_gpu_ids",
]
