"""GPU-side bench guards: hipGraph capture path stays functional."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_hip_graphs_capture():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--launch",
         "engine", "--hip-graphs", "--gpus", "1", "--steps", "3",
         "--warmup", "2", "--batch-size", "16", "--image-size", "64",
         "--num-classes", "10"],
        capture_output=True, text=True, timeout=420, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "hipGraph capture OK" in out.stdout, out.stdout[-500:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.strip().startswith("{") and '"metric"' in ln][-1]
    j = json.loads(line)
    assert j["value"] > 0
