"""Dump permlane{16,32}_swap semantics: a=0x1000+lane, b=0x2000+lane."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()
out = ext.permlane_swap_probe().cpu().numpy()
names = ["pl32.r0", "pl32.r1", "pl16.r0", "pl16.r1"]
for row, name in enumerate(names):
    desc = []
    for lane in range(64):
        v = out[row][lane]
        reg = "a" if (v >> 12) == 1 else "b"
        desc.append(f"{reg}{v & 0xFFF:02d}")
    print(name, " ".join(desc))
