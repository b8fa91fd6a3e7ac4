from .ray_launcher import RayLauncher
from .utils import _RayOutput, find_free_port, get_executable_cls

__all__ = ["RayLauncher", "_RayOutput", "find_free_port",
           "get_executable_cls"]
