"""Guards the driver's bench.py contract: flag surface, JSON schema,
and all three launch modes (actor fan-out, torchrun-external, raw
engine) on CPU."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def _run(args, timeout=420, env_extra=None):
    env = dict(os.environ)
    if env_extra:
        env.update(env_extra)
    out = subprocess.run([sys.executable] + args, capture_output=True,
                         text=True, timeout=timeout, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.strip().startswith("{") and '"metric"' in ln][-1]
    return json.loads(line)


def _check_contract(j, n_gpus):
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in j, f"missing {key}"
    assert j["n_gpus"] == n_gpus
    assert j["data"] == "synthetic"
    assert j["dtype"] == "bf16"
    assert j["scaling"] == "weak"
    assert j["higher_is_better"] is True
    assert j["value"] > 0
    assert "peak_mem_mib" in j["config"]


def test_bench_actor_mode_default():
    """The driver's N=1 shape: plain `python bench.py --gpus 1` drives
    Trainer+RayStrategy through the actor launcher."""
    j = _run([BENCH, "--gpus", "1", "--steps", "2", "--warmup", "1",
              "--batch-size", "4", "--image-size", "64",
              "--num-classes", "10"])
    _check_contract(j, 1)
    assert j["config"]["launch"] == "actor"
    assert j["config"]["parallelism"] == "dp1"


def test_bench_external_mode_torchrun():
    """The driver's N>1 shape: torchrun, one rank per GPU, external
    strategy mode."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", BENCH, "--gpus", "2", "--steps",
         "2", "--warmup", "1", "--batch-size", "4", "--image-size",
         "64", "--num-classes", "10"],
        capture_output=True, text=True, timeout=420, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.strip().startswith("{") and '"metric"' in ln][-1]
    j = json.loads(line)
    _check_contract(j, 2)
    assert j["config"]["launch"] == "external"
    assert j["config"]["parallelism"] == "dp2"


def test_bench_engine_mode():
    j = _run([BENCH, "--launch", "engine", "--gpus", "1", "--steps",
              "2", "--warmup", "1", "--batch-size", "4",
              "--image-size", "64", "--num-classes", "10"])
    _check_contract(j, 1)
    assert j["config"]["launch"] == "engine"


def test_bench_gpt2_sharded_labels():
    j = _run([BENCH, "--launch", "engine", "--model", "gpt2",
              "--steps", "1", "--warmup", "0", "--batch-size", "1",
              "--seq-len", "128"], timeout=600)
    _check_contract(j, 1)
    assert "sharded" in j["config"]["parallelism"]
    assert "tokens_per_s" in j["config"]


def test_sweep_harness_cpu():
    """scripts/sweep_buckets.py produces a table (guards the 8-GPU
    tuning harness the first multi-GPU node will use)."""
    out_md = os.path.join(REPO, "profiles", "_sweep_test_tmp.md")
    try:
        r = subprocess.run(
            [sys.executable, os.path.join(REPO, "scripts",
                                          "sweep_buckets.py"),
             "--gpus", "1", "--steps", "1", "--warmup", "0",
             "--batch-size", "4", "--buckets", "25", "--dtypes",
             "none", "--out", out_md],
            capture_output=True, text=True, timeout=420, cwd=REPO)
        assert r.returncode == 0, r.stderr[-1500:]
        body = open(out_md).read()
        assert "| bucket MB |" in body and "25.0" in body
    finally:
        if os.path.exists(out_md):
            os.remove(out_md)
