"""Tune hipBLASLt algorithm selection (PyTorch TunableOp) for the
GPT-2-XL bench shapes and write the pinned CSV the bench loads.

Run on an MI355X box:  python scripts/tune_gemms.py [--batch 16]
Writes ray_lightning_amd/ops/tunableop_gpt2xl.csv (commit it).
"""
import argparse
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
OUT = os.path.join(REPO, "ray_lightning_amd", "ops",
                   "tunableop_gpt2xl.csv")

p = argparse.ArgumentParser()
p.add_argument("--batch", type=int, default=16)
p.add_argument("--seq-len", type=int, default=1024)
p.add_argument("--steps", type=int, default=3)
args = p.parse_args()

os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = OUT
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "200")

import torch  # noqa: E402

sys.path.insert(0, REPO)
from ray_lightning_amd.models.gpt2 import (GPT2, GPT2Config,  # noqa: E402
                                           to_bf16_training)
from ray_lightning_amd.optim import ShardedFusedAdam  # noqa: E402

cfg = GPT2Config.gpt2_xl()
model = to_bf16_training(GPT2(cfg).cuda())
model.train()
opt = ShardedFusedAdam(model.parameters(), lr=1e-4)
x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq_len),
                  device="cuda")
y = torch.randint(0, cfg.vocab_size, (args.batch, args.seq_len),
                  device="cuda")
for i in range(args.steps):
    _, loss = model(x, y)
    loss.backward()
    opt.step()
    opt.zero_grad(set_to_none=True)
    torch.cuda.synchronize()
    print(f"step {i} loss {float(loss):.3f}", flush=True)
# torch writes the CSV at exit
print("tuning done ->", OUT)
