#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 data-parallel training throughput.

BASELINE.json metric: samples/sec (whole node), ResNet-50, RayStrategy
DDP at 1/2/4/8 MI355X workers, synthetic ImageNet-shaped data,
random-init weights, bf16 autocast compute.

Launch shapes:
  python bench.py --gpus 1 --steps K --warmup W            (single rank)
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...        (one rank/GPU)

The timed region is the full training step: forward (bf16 autocast),
backward with the NativeDDP bucketed RCCL all-reduce overlapped, bucket
finalize + 1/world scale, fused-SGD optimizer step. Data is pre-staged
synthetic batches (rotated) as declared in the output's "data" field.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

# MIOpen find mode: the default (DYNAMIC_HYBRID) benchmarks dynamic
# igemm solvers with real workspace during the *untimed* warmup steps.
# FAST (immediate heuristics) picks catastrophic no-workspace wrw-conv
# fallback kernels on MI355X (profiles/r01_resnet50_1gpu_fastfind.md).
# Override via env if needed.

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ray_lightning_amd.engine.comm import (TorchDistCommunicator,
                                           init_control_plane)
from ray_lightning_amd.engine.ddp import NativeDDP
from ray_lightning_amd.models.resnet import resnet50
from ray_lightning_amd.optim import FusedSGD


def build_comm(rank: int, world: int, device: torch.device):
    """Control plane (gloo) + data plane (native RCCL ext, else
    torch-dist RCCL)."""
    init_control_plane(rank, world)
    control = TorchDistCommunicator()
    if device.type == "cuda":
        from ray_lightning_amd.engine.rccl import (NativeRcclCommunicator,
                                                   rccl_available)
        if rccl_available():
            return control, NativeRcclCommunicator(control, device)
        import torch.distributed as dist
        return control, TorchDistCommunicator(
            dist.new_group(backend="nccl"))
    return control, control


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=256,
                   help="per-GPU batch (weak scaling)")
    p.add_argument("--bucket-mb", type=float, default=50.0)
    p.add_argument("--compression", choices=["none", "bf16"],
                   default="none", help="gradient comm dtype")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--memory-format", choices=["channels_last", "nchw"],
                   default="channels_last",
                   help="NHWC is the native MIOpen/CDNA4 conv layout")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    on_gpu = torch.cuda.is_available()

    if on_gpu:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)
        batch = args.batch_size
    else:
        # debug-only CPU path (the judged runs are on MI355X)
        device = torch.device("cpu")
        batch = min(args.batch_size, 8)

    torch.manual_seed(1234 + rank)
    # benchmark=True would force MIOpen full Find (minutes, cache-cold)
    torch.backends.cudnn.benchmark = False

    control = data_comm = None
    if world > 1:
        control, data_comm = build_comm(rank, world, device)

    model = resnet50(args.num_classes).to(device)
    channels_last = args.memory_format == "channels_last" and on_gpu
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    model.train()
    if world > 1:
        comm_dtype = torch.bfloat16 if args.compression == "bf16" else None
        wrapped = NativeDDP(model, data_comm,
                            bucket_cap_mb=args.bucket_mb,
                            comm_dtype=comm_dtype)
    else:
        wrapped = model
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                   weight_decay=1e-4)

    # pre-staged synthetic batches (rotate to defeat caching)
    n_buf = 4
    images = [torch.randn(batch, 3, 224, 224, device=device)
              .to(memory_format=torch.channels_last) if channels_last
              else torch.randn(batch, 3, 224, 224, device=device)
              for _ in range(n_buf)]
    labels = [torch.randint(0, args.num_classes, (batch,), device=device)
              for _ in range(n_buf)]

    autocast = (torch.autocast("cuda", dtype=torch.bfloat16)
                if on_gpu else torch.autocast("cpu", torch.bfloat16))

    def step(i: int) -> None:
        x, y = images[i % n_buf], labels[i % n_buf]
        with autocast:
            logits = model(x)
            loss = torch.nn.functional.cross_entropy(logits.float(), y)
        loss.backward()
        if world > 1:
            wrapped.finalize_backward()
        opt.step()
        opt.zero_grad(set_to_none=True)

    for i in range(args.warmup):
        step(i)

    if control is not None:
        control.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i + args.warmup)
    if on_gpu:
        torch.cuda.synchronize()
    if control is not None:
        control.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if control is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        control.all_reduce_(t, op="max")
        elapsed = float(t[0])

    if rank == 0:
        samples = world * batch * args.steps
        value = samples / elapsed
        out = {
            "metric": "samples/sec (whole node) ResNet-50 RayStrategy "
                      "DDP at 1/2/4/8 MI355X workers",
            "value": round(value, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "resnet50",
                "global_batch": world * batch,
                "image_size": 224,
                "parallelism": f"dp{world}",
                "grad_comm_dtype": args.compression,
                "bucket_cap_mb": args.bucket_mb,
                "memory_format": args.memory_format,
                "device": "cuda" if on_gpu else "cpu-debug",
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
