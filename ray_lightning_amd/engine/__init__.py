from .comm import (Communicator, TorchDistCommunicator,
                   destroy_control_plane, init_control_plane)
from .ddp import NativeDDP
from .sharded import ShardedDDP, ShardedOptimizer

__all__ = [
    "Communicator", "TorchDistCommunicator", "destroy_control_plane",
    "init_control_plane", "NativeDDP", "ShardedDDP", "ShardedOptimizer",
]
