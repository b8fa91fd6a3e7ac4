from .base import SingleDeviceStrategy, Strategy
from .ray_ddp import RayStrategy
from .ray_ddp_sharded import RayShardedStrategy
from .ray_horovod import HorovodRayStrategy

STRATEGY_REGISTRY = {
    "ddp_ray": RayStrategy,
    "ddp_sharded_ray": RayShardedStrategy,
    "horovod_ray": HorovodRayStrategy,
    "single_device": SingleDeviceStrategy,
}

__all__ = [
    "Strategy", "SingleDeviceStrategy", "RayStrategy",
    "RayShardedStrategy", "HorovodRayStrategy", "STRATEGY_REGISTRY",
]
