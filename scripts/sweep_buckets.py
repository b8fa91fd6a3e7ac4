#!/usr/bin/env python3
"""Bucket-size / gradient-compression sweep for the DDP engine.

One command -> one markdown table in ``profiles/`` backing the
"bucket sizes tuned for 7-link xGMI" claim with measurements
(VERDICT r01 weak#4): runs the flagship bench over
bucket_cap x comm_dtype at N GPUs and tabulates samples/s.

    python scripts/sweep_buckets.py --gpus 8 --steps 20 --warmup 8
    python scripts/sweep_buckets.py --gpus 8 --model gpt2-xl \
        --strategy sharded --buckets 12,25,50,100

N>1 launches each cell under ``torch.distributed.run`` (one rank per
GPU, external mode — same path the driver's scale runs use); N=1 runs
the bench directly. Each cell is an independent process pair so MIOpen
/ hipBLASLt caches stay warm across cells on the same box.
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cell(args, bucket_mb: float, compression: str) -> dict:
    cmd = [sys.executable]
    if args.gpus > 1:
        cmd += ["-m", "torch.distributed.run", "--nnodes=1",
                f"--nproc-per-node={args.gpus}",
                "--master-addr", "127.0.0.1",
                "--master-port", str(29600 + (os.getpid() % 200))]
    cmd += [os.path.join(REPO, "bench.py"),
            "--gpus", str(args.gpus),
            "--steps", str(args.steps),
            "--warmup", str(args.warmup),
            "--model", args.model,
            "--strategy", args.strategy,
            "--bucket-mb", str(bucket_mb),
            "--compression", compression]
    if args.batch_size:
        cmd += ["--batch-size", str(args.batch_size)]
    t0 = time.time()
    out = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=args.cell_timeout, cwd=REPO)
    wall = time.time() - t0
    line = None
    for ln in out.stdout.splitlines():
        ln = ln.strip()
        if ln.startswith("{") and '"metric"' in ln:
            line = json.loads(ln)
    if line is None:
        return {"bucket_mb": bucket_mb, "compression": compression,
                "error": (out.stderr or out.stdout)[-500:], "wall_s": wall}
    line_out = {"bucket_mb": bucket_mb, "compression": compression,
                "samples_per_s": line["value"],
                "ms_per_step": line["ms_per_step"],
                "peak_mem_mib": line["config"].get("peak_mem_mib"),
                "wall_s": round(wall, 1)}
    return line_out


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--strategy", default="ddp")
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--buckets", default="12,25,50,100",
                   help="comma-separated bucket caps in MB")
    p.add_argument("--dtypes", default="none,bf16",
                   help="comma-separated comm dtypes (none=fp32)")
    p.add_argument("--cell-timeout", type=int, default=900)
    p.add_argument("--out", default=None,
                   help="markdown output path (default "
                        "profiles/bucket_sweep_<model>_<N>gpu.md)")
    args = p.parse_args()
    if args.strategy == "sharded":
        args.dtypes = "none"  # sharded engine comms in the grad dtype

    buckets = [float(b) for b in args.buckets.split(",")]
    dtypes = [d.strip() for d in args.dtypes.split(",") if d.strip()]

    rows = []
    for comp in dtypes:
        for b in buckets:
            print(f"[sweep] bucket={b} MB comm={comp} ...", flush=True)
            r = run_cell(args, b, comp)
            rows.append(r)
            print(f"[sweep]   -> {r}", flush=True)

    out_path = args.out or os.path.join(
        REPO, "profiles",
        f"bucket_sweep_{args.model}_{args.gpus}gpu.md")
    best = max((r for r in rows if "error" not in r),
               default=None, key=lambda r: r["samples_per_s"])
    with open(out_path, "w") as f:
        f.write(f"# Bucket-size / compression sweep — {args.model} "
                f"{args.strategy} @ {args.gpus} GPU(s)\n\n")
        f.write(f"steps={args.steps} warmup={args.warmup} "
                f"batch={args.batch_size or 'default'} "
                f"(weak scaling, one rank per GPU over RCCL/xGMI)\n\n")
        f.write("| bucket MB | comm dtype | samples/s | ms/step | "
                "peak MiB |\n|---|---|---|---|---|\n")
        for r in rows:
            if "error" in r:
                f.write(f"| {r['bucket_mb']} | {r['compression']} | "
                        f"ERROR | — | — |\n")
            else:
                mark = " **best**" if r is best else ""
                f.write(f"| {r['bucket_mb']} | {r['compression']} | "
                        f"{r['samples_per_s']}{mark} | "
                        f"{r['ms_per_step']} | {r['peak_mem_mib']} |\n")
        errs = [r for r in rows if "error" in r]
        for r in errs:
            f.write(f"\n- error @ bucket={r['bucket_mb']} "
                    f"comm={r['compression']}: `{r['error'][:300]}`\n")
    print(f"[sweep] wrote {out_path}")


if __name__ == "__main__":
    main()
