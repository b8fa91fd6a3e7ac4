"""RayStrategy: data-parallel training over actor workers.

API-parity re-implementation of the reference's flagship strategy
(reference ray_ddp.py:23-333) on the MI355X-native stack: constructor
signature, rank bookkeeping, device resolution, sampler kwargs and
teardown semantics match; the substance underneath is this framework's
own actor runtime, Gloo control plane and native-RCCL/NativeDDP gradient
engine instead of Ray + torch DDP/NCCL.

Also supports **external launch mode**: when RANK/WORLD_SIZE env vars are
already set (e.g. `torch.distributed.run` on a GPU node), the strategy
skips the actor launcher and runs the trainer in-process as that rank —
one process per GPU over RCCL, the canonical MI355X deployment shape.
"""
from __future__ import annotations

import os
import warnings
from typing import Any, Callable, Dict, List, Optional, Tuple, Union

import torch

from ..engine.comm import (Communicator, TorchDistCommunicator,
                           destroy_control_plane, init_control_plane)
from ..engine.ddp import NativeDDP
from ..launchers.ray_launcher import RayLauncher
from ..runtime.actor import get_gpu_ids
from .base import Strategy


class RayStrategy(Strategy):
    """Data-parallel strategy over this framework's worker actors
    (ctor parity: reference ray_ddp.py:69-116)."""

    strategy_name = "ddp_ray"
    is_remote_launch = True

    def __init__(self,
                 num_workers: int = 1,
                 num_cpus_per_worker: int = 1,
                 use_gpu: bool = False,
                 init_hook: Optional[Callable] = None,
                 resources_per_worker: Optional[Dict] = None,
                 **ddp_kwargs: Union[Any, Dict[str, Any]]):
        super().__init__()
        resources_per_worker = dict(resources_per_worker or {})
        self.nickname = "ddp_ray"
        self.num_workers = int(num_workers)
        self.num_cpus_per_worker = resources_per_worker.pop(
            "CPU", num_cpus_per_worker)
        if "GPU" in resources_per_worker:
            self.num_gpus_per_worker = resources_per_worker.pop("GPU")
        else:
            self.num_gpus_per_worker = int(use_gpu)
        self.use_gpu = self.num_gpus_per_worker > 0

        if self.use_gpu and self.num_gpus_per_worker < 1 and num_workers > 1:
            warnings.warn(
                "Identified less than 1 GPU per worker: RCCL cannot share "
                "one device between ranks; gradient communication falls "
                "back to the Gloo data plane "
                "(set PL_TORCH_DISTRIBUTED_BACKEND=gloo to silence).")

        self.additional_resources_per_worker = resources_per_worker
        self.init_hook = init_hook
        self._ddp_kwargs = ddp_kwargs

        self._local_rank = 0
        self._global_rank = 0
        self._node_rank = 0
        self._is_remote = False
        self._device: Optional[torch.device] = None
        self._global_to_local: List[Tuple[int, int]] = []
        self._comm: Optional[Communicator] = None
        self._data_comm: Optional[Communicator] = None
        self._external_mode = False

    # ------------------------------------------------------------------ #
    # launcher wiring
    # ------------------------------------------------------------------ #
    def _configure_launcher(self) -> None:
        self._launcher = RayLauncher(self)

    def connect(self, trainer) -> None:
        super().connect(trainer)
        # external (torchrun) mode: ranks pre-assigned by the launcher env
        if not self._is_remote and "RANK" in os.environ \
                and "WORLD_SIZE" in os.environ:
            self._external_mode = True

    @property
    def is_remote_launch(self) -> bool:  # type: ignore[override]
        return not self._external_mode

    @property
    def accelerator(self) -> str:
        """"_gpu" defers device binding to the worker (the driver may be
        CPU-only) — reference ray_ddp.py:112-113."""
        return "_gpu" if self.use_gpu else "cpu"

    def set_remote(self, remote: bool) -> None:
        self._is_remote = remote

    def set_global_to_local(self,
                            global_to_local: List[Tuple[int, int]]) -> None:
        self._global_to_local = list(global_to_local)

    # ------------------------------------------------------------------ #
    # rank bookkeeping (reference ray_ddp.py:145-257)
    # ------------------------------------------------------------------ #
    def set_world_ranks(self, process_idx: int = 0) -> None:
        if not self._is_remote and not self._external_mode:
            return  # driver side: launcher owns ranks
        self._global_rank = process_idx
        if self._global_to_local:
            self._local_rank, self._node_rank = \
                self._global_to_local[self._global_rank]
        else:
            self._local_rank, self._node_rank = process_idx, 0

    @property
    def world_size(self) -> int:
        if self._external_mode:
            return int(os.environ.get("WORLD_SIZE", "1"))
        return self.num_workers

    @property
    def global_rank(self) -> int:
        return self._global_rank

    @property
    def local_rank(self) -> int:
        return self._local_rank

    @property
    def node_rank(self) -> int:
        return self._node_rank

    @property
    def distributed_sampler_kwargs(self) -> Optional[Dict[str, int]]:
        if self.world_size <= 1:
            return None
        return {"num_replicas": self.world_size, "rank": self.global_rank}

    # ------------------------------------------------------------------ #
    # device resolution (reference ray_ddp.py:259-313)
    # ------------------------------------------------------------------ #
    @property
    def root_device(self) -> torch.device:
        if self._device is not None:
            return self._device
        if self._external_mode and self.use_gpu:
            local = int(os.environ.get("LOCAL_RANK", "0"))
            # modulo lets N ranks share fewer devices (gloo-data-plane
            # validation on small boxes); an 8-GPU deployment maps 1:1
            if torch.cuda.is_available():
                local %= max(1, torch.cuda.device_count())
            return torch.device("cuda", local)
        if self._is_remote and self.use_gpu and torch.cuda.is_available():
            gpu_ids = get_gpu_ids()
            if gpu_ids:
                # Physical id -> index within this process's visible set
                # (HIP_VISIBLE_DEVICES union written by the launcher,
                #  reference ray_ddp.py:271-302).
                visible = os.environ.get(
                    "HIP_VISIBLE_DEVICES",
                    os.environ.get("CUDA_VISIBLE_DEVICES"))
                if visible:
                    vis = [int(x) for x in visible.split(",")]
                    idx = vis.index(gpu_ids[0])
                else:
                    idx = gpu_ids[0]
                return torch.device("cuda", idx)
            return torch.device("cuda", 0)
        return torch.device("cpu")

    @root_device.setter
    def root_device(self, device) -> None:
        self._device = torch.device(device) if device is not None else None

    # ------------------------------------------------------------------ #
    # worker-side setup (reference ray_ddp.py:161-203)
    # ------------------------------------------------------------------ #
    def _worker_setup(self, process_idx: int) -> None:
        from ..trainer.trainer import _reset_seed
        from ..util import rank_zero_only
        _reset_seed()
        self.set_world_ranks(process_idx)
        # "rank-0-only" helpers must act on the ACTOR's rank
        # (reference ray_ddp.py:169)
        rank_zero_only.rank = self.global_rank
        if self.world_size > 1:
            init_control_plane(self.global_rank, self.world_size)
            self._comm = TorchDistCommunicator()
        self._setup_data_plane()
        if self.root_device.type == "cuda":
            torch.cuda.set_device(self.root_device)

    def _setup_data_plane(self) -> None:
        """Choose the gradient-communication path.

        GPU with whole devices -> native RCCL communicator (ncclUniqueId
        broadcast over the control plane — SURVEY.md N1). Fractional GPU
        or CPU -> the Gloo control plane doubles as data plane
        (reference ray_ddp.py:91-100). PL_TORCH_DISTRIBUTED_BACKEND=gloo
        forces the fallback."""
        self._data_comm = self._comm
        if self.world_size <= 1:
            return
        backend_override = os.environ.get("PL_TORCH_DISTRIBUTED_BACKEND")
        whole_gpus = self.num_gpus_per_worker >= 1 or self._external_mode
        if self._external_mode and torch.cuda.is_available():
            # ranks sharing one device (validation boxes with fewer
            # GPUs than ranks) cannot run RCCL — fall back to gloo
            # instead of hanging in ncclCommInitRank
            local_world = int(os.environ.get(
                "LOCAL_WORLD_SIZE", os.environ.get("WORLD_SIZE", "1")))
            if local_world > torch.cuda.device_count():
                whole_gpus = False
        if self.use_gpu and whole_gpus and torch.cuda.is_available() \
                and backend_override != "gloo":
            from ..engine.rccl import NativeRcclCommunicator, rccl_available
            native_ok = False
            if rccl_available():
                try:
                    self._data_comm = NativeRcclCommunicator(
                        self._comm, device=self.root_device)
                    native_ok = True
                except Exception:  # noqa: BLE001 — fall back below
                    native_ok = False
            if not native_ok:
                # torch-dist RCCL fallback: still RCCL over xGMI, managed
                # by torch instead of our extension.
                import torch.distributed as dist
                nccl_group = dist.new_group(backend="nccl")
                self._data_comm = TorchDistCommunicator(nccl_group)

    # external (torchrun) mode initialization happens lazily here
    def setup_environment(self) -> None:
        if self._external_mode and self._comm is None:
            rank = int(os.environ["RANK"])
            world = int(os.environ["WORLD_SIZE"])
            local = int(os.environ.get("LOCAL_RANK", "0"))
            self._global_rank, self._local_rank = rank, local
            if world > 1:
                init_control_plane(rank, world)
                self._comm = TorchDistCommunicator()
            self._setup_data_plane()
            if self.use_gpu and torch.cuda.is_available():
                torch.cuda.set_device(self.root_device)

    # ------------------------------------------------------------------ #
    # model wrapping + collectives
    # ------------------------------------------------------------------ #
    def wrap_model(self, model: torch.nn.Module) -> torch.nn.Module:
        if self.world_size <= 1 or self._data_comm is None:
            return model
        return NativeDDP(model, self._data_comm, **self._ddp_kwargs)

    def reduce(self, tensor: torch.Tensor, op: str = "mean"):
        if self._comm is None:
            return tensor
        self._comm.all_reduce_(tensor, op=op)
        return tensor

    def barrier(self) -> None:
        if self._comm is not None:
            self._comm.barrier()

    def broadcast_object(self, obj: Any, src: int = 0) -> Any:
        if self._comm is None:
            return obj
        return self._comm.broadcast_object(obj, src=src)

    # ------------------------------------------------------------------ #
    def teardown_worker(self) -> None:
        """Worker-side cleanup after one launch."""
        if self._data_comm is not None and self._data_comm is not self._comm:
            try:
                self._data_comm.synchronize()
            except Exception:
                pass
        destroy_control_plane()
        self._comm = None
        self._data_comm = None

    def teardown(self) -> None:
        """Driver-side teardown (reference ray_ddp.py:326-333)."""
        if self._launcher is not None:
            self._launcher.teardown_workers()
