"""Dump ds_read_b64_tr_b16 element maps for several address patterns."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()
for (a, b, c) in [(8, 0, 0), (0, 8, 128), (0, 8, 256)]:
    out = ext.tr16_probe(a, b, c).cpu().numpy()
    print(f"--- addr = {a}*lane + {b}*(lane&15) + {c}*(lane>>4)")
    for lane in range(0, 64):
        print(f"l{lane:02d}: " + " ".join(f"{v:4d}" for v in out[lane]),
              end="   " if lane % 2 == 0 else "\n")
