"""External (torchrun) strategy mode: the full Trainer fit/validate
path per rank without the actor launcher — the N>1 deployment shape
the driver uses, on CPU/gloo."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = r"""
import os
import sys

sys.path.insert(0, %(repo)r)
sys.path.insert(0, %(tests)r)
import torch
from utils import BoringModel
from ray_lightning_amd import RayStrategy, Trainer

rank = int(os.environ["RANK"])
model = BoringModel()
trainer = Trainer(max_epochs=2, limit_train_batches=4,
                  limit_val_batches=2, num_sanity_val_steps=0,
                  enable_checkpointing=False,
                  strategy=RayStrategy(num_workers=2))
trainer.fit(model)
assert trainer.state.finished
assert trainer.strategy._external_mode
assert trainer.world_size == 2
assert trainer.global_rank == rank
# validate() reuses the initialized external environment
out = trainer.validate(model)
print(f"RANK{rank}-OK", flush=True)
"""


def test_external_mode_fit_and_validate(tmp_path):
    script = SCRIPT % {"repo": REPO,
                       "tests": os.path.join(REPO, "tests")}
    sp = str(tmp_path / "ext.py")
    with open(sp, "w") as f:
        f.write(script)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29551", sp],
        capture_output=True, text=True, timeout=420, cwd=str(tmp_path))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "RANK0-OK" in out.stdout
    assert "RANK1-OK" in out.stdout
