"""Examples run end-to-end with --smoke-test (the reference CI runs its
examples the same way, .github/workflows/test.yaml:95-107)."""
import os
import subprocess
import sys

import pytest

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_EX = os.path.join(_ROOT, "ray_lightning_amd", "examples")


@pytest.mark.parametrize("script", [
    "ray_ddp_example.py",
    "ray_ddp_tune.py",
    "ray_horovod_example.py",
    "ray_ddp_sharded_example.py",
])
def test_example_smoke(script):
    proc = subprocess.run(
        [sys.executable, os.path.join(_EX, script), "--smoke-test"],
        capture_output=True, text=True, timeout=600, cwd=_ROOT)
    assert proc.returncode == 0, proc.stderr[-2000:]
    assert "smoke OK" in proc.stdout
