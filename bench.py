#!/usr/bin/env python3
"""Flagship benchmark: data-parallel training throughput on MI355X.

BASELINE.json metric (default): samples/sec (whole node), ResNet-50,
RayStrategy DDP at 1/2/4/8 MI355X workers, synthetic ImageNet-shaped
data, random-init weights, bf16 autocast compute.

Also runs BASELINE config 4 via ``--model gpt2-xl --strategy sharded``:
GPT-2-XL bf16 weights, ShardedOptimizer (OSS 1/N optimizer state) +
ShardedDDP reduce-to-owner gradients + sharded fused-Adam HIP kernel.

Launch shapes (the measured path is ALWAYS Trainer + RayStrategy +
native RCCL — the claimed BASELINE config, reference
launchers/ray_launcher.py:221-250 fan-out included where possible):

  python bench.py --gpus N --steps K --warmup W
      single driver process; ``Trainer(strategy=RayStrategy(
      num_workers=N, use_gpu=True)).fit(...)`` spawns N worker actors
      through RayLauncher (HIP_VISIBLE_DEVICES union, object-store model
      ship, rank-0 collect) — "launch": "actor".

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...
      one externally-launched rank per GPU; each rank runs the SAME
      Trainer+RayStrategy stack in external mode (strategy detects
      RANK/WORLD_SIZE, skips only the actor spawn) — "launch":
      "external". This is the shape the driver uses for N>1.

  python bench.py --launch engine ...
      raw engine loop (no Trainer/strategy) — kept to measure
      orchestration overhead against; "launch": "engine".

The timed region is identical in all modes: W untimed warmup steps,
then exactly K full training steps (forward, backward with the bucketed
RCCL collective overlapped, bucket finalize + 1/world scale, fused
optimizer step) bracketed by barrier + ``torch.cuda.synchronize()``,
MAX over ranks. Peak GPU memory over the timed region is reported as
``peak_mem_mib`` (max over ranks).
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
# fallback kernels on MI355X: 212 vs 6014 samples/s measured
# (profiles/r01_resnet50_1gpu_fastfind.md). Override via env if needed.

# Drop the naive reference convolutions from MIOpen's find candidate
# list: at batch 768 each naive wrw timing costs ~0.7 s and the find
# phase spends ~3 GPU-minutes on them (profiles/prof9) while never
# winning a shape. The igemm/CK solvers are unaffected.
os.environ.setdefault("MIOPEN_DEBUG_CONV_DIRECT_NAIVE_CONV_FWD", "0")
os.environ.setdefault("MIOPEN_DEBUG_CONV_DIRECT_NAIVE_CONV_BWD", "0")
os.environ.setdefault("MIOPEN_DEBUG_CONV_DIRECT_NAIVE_CONV_WRW", "0")

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args() -> argparse.Namespace:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", choices=["resnet50", "gpt2", "gpt2-xl"],
                   default="resnet50")
    p.add_argument("--strategy", choices=["ddp", "sharded"], default=None,
                   help="default: ddp for resnet50, sharded for gpt2*")
    p.add_argument("--launch", choices=["auto", "actor", "external",
                                        "engine"], default="auto",
                   help="auto = external when torchrun env is present, "
                        "else actor (RayLauncher fan-out)")
    p.add_argument("--batch-size", type=int, default=None,
                   help="per-GPU batch (weak scaling); defaults 768 "
                        "resnet / 16 gpt2 (measured best on MI355X: "
                        "256->512->768 = 7.9k->8.7k->9.0k samples/s)")
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--bucket-mb", type=float, default=50.0)
    p.add_argument("--compression", choices=["none", "bf16"],
                   default="none", help="gradient comm dtype (ddp)")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--memory-format", choices=["channels_last", "nchw"],
                   default="channels_last",
                   help="NHWC is the native MIOpen/CDNA4 conv layout")
    p.add_argument("--hip-graphs", action="store_true",
                   help="capture the train step in hipGraphs (engine "
                        "mode, world=1): one replay per step instead "
                        "of ~800 kernel launches")
    p.add_argument("--nccl-debug", action="store_true",
                   help="print per-rank RCCL ring/channel topology "
                        "(NCCL_DEBUG=INFO) so multi-GPU runs are "
                        "diagnosable")
    args = p.parse_args()

    is_gpt = args.model.startswith("gpt2")
    if args.strategy is None:
        args.strategy = "sharded" if is_gpt else "ddp"
    if args.batch_size is None:
        args.batch_size = 16 if is_gpt else 768
    if args.launch == "auto":
        args.launch = ("external" if "WORLD_SIZE" in os.environ
                       else "actor")
    if args.nccl_debug:
        os.environ.setdefault("NCCL_DEBUG", "INFO")
        os.environ.setdefault("NCCL_DEBUG_SUBSYS", "INIT,GRAPH,TUNING")
    if is_gpt:
        # hipBLASLt algo pinning: load the TunableOp results tuned on
        # MI355X for the GPT-2-XL shapes (scripts/tune_gemms.py wrote
        # them; read-only here — no tuning cost in the bench).
        csvp = os.path.join(
            os.path.dirname(os.path.abspath(__file__)),
            "ray_lightning_amd", "ops", "tunableop_gpt2xl.csv")
        if os.path.exists(csvp):
            os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
            os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
            os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", csvp)
    return args


def emit(args, world: int, elapsed: float, peak_mem_mib: float,
         launch: str) -> None:
    """Rank-0/driver JSON line — the driver contract."""
    is_gpt = args.model.startswith("gpt2")
    batch = args.batch_size
    samples = world * batch * args.steps
    value = samples / elapsed
    cfg_out = {
        "model": args.model,
        "global_batch": world * batch,
        "parallelism": (f"dp{world}" if args.strategy == "ddp"
                        else f"sharded-dp{world}"),
        "launch": launch,
        "bucket_cap_mb": args.bucket_mb,
        "peak_mem_mib": round(peak_mem_mib, 1),
        "device": "cuda" if torch.cuda.is_available() else "cpu-debug",
    }
    if is_gpt:
        cfg_out["seq_len"] = args.seq_len
        cfg_out["tokens_per_s"] = round(value * args.seq_len, 1)
        cfg_out["optimizer"] = "sharded-fused-adamw(bf16+fp32 master)"
    else:
        cfg_out["image_size"] = args.image_size
        cfg_out["grad_comm_dtype"] = args.compression
        cfg_out["memory_format"] = args.memory_format
    out = {
        "metric": "samples/sec (whole node) ResNet-50 RayStrategy "
                  "DDP at 1/2/4/8 MI355X workers" if not is_gpt else
                  f"samples/sec (whole node) {args.model} "
                  f"RayShardedStrategy",
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
        "config": cfg_out,
    }
    print(json.dumps(out), flush=True)


# ---------------------------------------------------------------------- #
# Trainer + RayStrategy modes (actor fan-out or torchrun-external)
# ---------------------------------------------------------------------- #
def run_trainer_mode(args) -> None:
    from ray_lightning_amd import RayShardedStrategy, RayStrategy
    from ray_lightning_amd.benchmarks import (BenchGPT2, BenchResNet,
                                              BenchTimerCallback)
    from ray_lightning_amd.trainer import Trainer

    on_gpu = torch.cuda.is_available()
    is_gpt = args.model.startswith("gpt2")
    external = args.launch == "external"
    if external and "WORLD_SIZE" not in os.environ:
        raise SystemExit("--launch external requires torchrun env "
                         "(RANK/WORLD_SIZE/MASTER_ADDR)")
    if not external:
        # the strategy treats a pre-set RANK as external mode; an actor
        # driver must not inherit one
        for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
            os.environ.pop(k, None)

    world = (int(os.environ["WORLD_SIZE"]) if external else args.gpus)
    rank = int(os.environ.get("RANK", "0"))
    batch = args.batch_size if on_gpu else min(args.batch_size, 8)
    args.batch_size = batch
    n_batches = args.warmup + args.steps

    torch.backends.cudnn.benchmark = False

    if is_gpt:
        module = BenchGPT2(model_name=args.model, batch_size=batch,
                           seq_len=args.seq_len, n_batches=n_batches)
        precision = 32  # bf16 weights already; autocast would only add
        # per-op cast traffic
    else:
        module = BenchResNet(
            batch_size=batch, num_classes=args.num_classes,
            image_size=args.image_size,
            channels_last=args.memory_format == "channels_last",
            n_batches=n_batches)
        precision = "bf16"

    strat_kwargs = dict(num_workers=world, use_gpu=on_gpu,
                        bucket_cap_mb=args.bucket_mb)
    if args.strategy == "sharded":
        strategy = RayShardedStrategy(**strat_kwargs)
    else:
        if args.compression == "bf16":
            strat_kwargs["comm_dtype"] = torch.bfloat16
        strategy = RayStrategy(**strat_kwargs)

    timer = BenchTimerCallback(warmup=args.warmup, steps=args.steps)
    trainer = Trainer(max_epochs=1, strategy=strategy, precision=precision,
                      callbacks=[timer], enable_checkpointing=False,
                      enable_progress_bar=False, num_sanity_val_steps=0,
                      log_every_n_steps=10 ** 9)
    trainer.fit(module)

    metrics = trainer.callback_metrics
    if "bench_elapsed_s" not in metrics:
        raise RuntimeError(
            "bench timer metrics did not come back through the "
            f"launcher collect protocol: {sorted(metrics)}")
    elapsed = float(metrics["bench_elapsed_s"])
    peak = float(metrics.get("bench_peak_mem_mib", 0.0))
    if rank == 0:
        emit(args, world, elapsed, peak, launch=args.launch)


# ---------------------------------------------------------------------- #
# raw engine mode (no Trainer/strategy) — orchestration-overhead control
# ---------------------------------------------------------------------- #
def build_comm(rank: int, world: int, device: torch.device):
    """Control plane (gloo) + data plane (native RCCL ext, else
    torch-dist RCCL)."""
    from ray_lightning_amd.engine.comm import (TorchDistCommunicator,
                                               init_control_plane)
    init_control_plane(rank, world)
    control = TorchDistCommunicator()
    # PL_TORCH_DISTRIBUTED_BACKEND=gloo forces the CPU data plane (the
    # fractional-GPU / ranks-sharing-one-device case, like the strategy)
    local_world = int(os.environ.get("LOCAL_WORLD_SIZE", str(world)))
    if device.type == "cuda" and \
            os.environ.get("PL_TORCH_DISTRIBUTED_BACKEND") != "gloo" \
            and local_world <= torch.cuda.device_count():
        from ray_lightning_amd.engine.rccl import (NativeRcclCommunicator,
                                                   rccl_available)
        if rccl_available():
            try:
                return control, NativeRcclCommunicator(control, device)
            except Exception as e:  # noqa: BLE001 — never strand an
                # 8-GPU run on an init quirk; torch-dist IS RCCL too
                if rank == 0:
                    print(f"[bench] native RCCL init failed ({e}); "
                          "falling back to torch-dist RCCL",
                          flush=True)
        import torch.distributed as dist
        return control, TorchDistCommunicator(
            dist.new_group(backend="nccl"))
    return control, control


def run_engine_mode(args) -> None:
    from ray_lightning_amd.engine.ddp import NativeDDP
    from ray_lightning_amd.engine.sharded import (ShardedDDP,
                                                  ShardedOptimizer)
    from ray_lightning_amd.models.gpt2 import GPT2, GPT2Config
    from ray_lightning_amd.models.resnet import resnet50
    from ray_lightning_amd.optim import FusedSGD, ShardedFusedAdam

    is_gpt = args.model.startswith("gpt2")
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    on_gpu = torch.cuda.is_available()

    if on_gpu:
        # modulo: lets N ranks share fewer devices (fractional-GPU /
        # gloo validation); the 8-GPU deployment maps 1:1
        device = torch.device("cuda",
                              local_rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
        batch = args.batch_size
    else:
        # debug-only CPU path (the judged runs are on MI355X)
        device = torch.device("cpu")
        batch = min(args.batch_size, 8)
    args.batch_size = batch

    torch.manual_seed(1234 + rank)
    # benchmark=True would force MIOpen exhaustive Find (minutes on a
    # cache-cold box); default find mode is already near-optimal.
    torch.backends.cudnn.benchmark = False

    control = data_comm = None
    if world > 1:
        control, data_comm = build_comm(rank, world, device)

    # model + optimizer + synthetic data
    if is_gpt:
        cfg = (GPT2Config.gpt2_xl() if args.model == "gpt2-xl"
               else GPT2Config.gpt2())
        cfg.n_positions = max(args.seq_len, 1024)
        from ray_lightning_amd.models.gpt2 import to_bf16_training
        model = to_bf16_training(GPT2(cfg).to(device))
        model.train()
        decay = [pm for pm in model.parameters() if pm.dim() >= 2]
        nodecay = [pm for pm in model.parameters() if pm.dim() < 2]
        opt = ShardedFusedAdam(
            [{"params": decay, "weight_decay": 0.1},
             {"params": nodecay, "weight_decay": 0.0}],
            lr=6e-4, betas=(0.9, 0.95))
        if world > 1:
            if args.strategy == "sharded":
                opt = ShardedOptimizer(opt, data_comm,
                                       bucket_cap_mb=args.bucket_mb)
                wrapped = ShardedDDP(model, data_comm, opt,
                                     bucket_cap_mb=args.bucket_mb)
            else:
                wrapped = NativeDDP(model, data_comm,
                                    bucket_cap_mb=args.bucket_mb)
        else:
            wrapped = model
        vocab = cfg.vocab_size
        n_buf = 4
        xs = [torch.randint(0, vocab, (batch, args.seq_len), device=device)
              for _ in range(n_buf)]
        ys = [torch.randint(0, vocab, (batch, args.seq_len), device=device)
              for _ in range(n_buf)]

        def fwd(i: int):
            _, loss = model(xs[i % n_buf], ys[i % n_buf])
            return loss
    else:
        model = resnet50(args.num_classes).to(device)
        channels_last = args.memory_format == "channels_last" and on_gpu
        if channels_last:
            model = model.to(memory_format=torch.channels_last)
        model.train()
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                       weight_decay=1e-4)
        if world > 1:
            if args.strategy == "sharded":
                opt = ShardedOptimizer(opt, data_comm,
                                       bucket_cap_mb=args.bucket_mb)
                wrapped = ShardedDDP(model, data_comm, opt,
                                     bucket_cap_mb=args.bucket_mb)
            else:
                comm_dtype = (torch.bfloat16 if args.compression == "bf16"
                              else None)
                wrapped = NativeDDP(model, data_comm,
                                    bucket_cap_mb=args.bucket_mb,
                                    comm_dtype=comm_dtype)
        else:
            wrapped = model
        n_buf = 4
        images = [torch.randn(batch, 3, args.image_size, args.image_size,
                              device=device)
                  .to(memory_format=torch.channels_last) if channels_last
                  else torch.randn(batch, 3, args.image_size,
                                   args.image_size, device=device)
                  for _ in range(n_buf)]
        labels = [torch.randint(0, args.num_classes, (batch,),
                                device=device) for _ in range(n_buf)]
        ac = (torch.autocast("cuda", dtype=torch.bfloat16) if on_gpu
              else torch.autocast("cpu", torch.bfloat16))

        def fwd(i: int):
            with ac:
                logits = model(images[i % n_buf])
                return torch.nn.functional.cross_entropy(
                    logits.float(), labels[i % n_buf])

    def step(i: int) -> None:
        loss = fwd(i)
        loss.backward()
        if world > 1:
            wrapped.finalize_backward()
        opt.step()
        opt.zero_grad(set_to_none=True)

    for i in range(args.warmup):
        step(i)

    if args.hip_graphs and on_gpu and world == 1:
        # capture one graph per rotating buffer (keeps the exact data
        # rotation) after the warmup has settled MIOpen/hipBLASLt algos
        try:
            torch.cuda.synchronize()
            graphs = []
            for i in range(n_buf):
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    step(i)
                graphs.append(g)
            torch.cuda.synchronize()
            eager_step = step

            def step(i):  # noqa: F811 — replay replaces the eager step
                graphs[i % n_buf].replay()

            if rank == 0:
                print("[bench] hipGraph capture OK "
                      f"({n_buf} graphs)", flush=True)
        except Exception as e:  # noqa: BLE001 — fall back to eager
            if rank == 0:
                print(f"[bench] hipGraph capture failed ({e}); "
                      "eager steps", flush=True)

    if control is not None:
        control.barrier()
    if on_gpu:
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i + args.warmup)
    if on_gpu:
        torch.cuda.synchronize()
    if control is not None:
        control.barrier()
    elapsed = time.perf_counter() - t0
    peak = (torch.cuda.max_memory_allocated() / (1024 * 1024)
            if on_gpu else 0.0)

    # max over ranks
    if control is not None:
        t = torch.tensor([elapsed, peak], dtype=torch.float64)
        control.all_reduce_(t, op="max")
        elapsed, peak = float(t[0]), float(t[1])

    if rank == 0:
        emit(args, world, elapsed, peak, launch="engine")


def main() -> None:
    args = parse_args()
    if args.launch == "engine":
        run_engine_mode(args)
    else:
        run_trainer_mode(args)


if __name__ == "__main__":
    main()
