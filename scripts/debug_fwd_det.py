"""Bisect dkv nondeterminism: permlane vs shfl path, per-tensor."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()
print("ext:", ops._hip_ops.__file__ if hasattr(ops, "_hip_ops") else "?")
torch.manual_seed(9)
B, H, T = 2, 3, 256
q = torch.randn(B, H, T, 64, device="cuda").bfloat16()
k = torch.randn_like(q)
v = torch.randn_like(q)
dy = torch.randn_like(q)

for pl in (True, False):
    o, lse = ext.flash_attn_fwd_v3(q, k, v, 0.125, pl)
    b0 = [t.clone() for t in
          ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, 0.125, pl)]
    bad = {"dq": 0, "dk": 0, "dv": 0}
    for i in range(15):
        bb = ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, 0.125, pl)
        for name, t, ref in zip(("dq", "dk", "dv"), bb, b0):
            if not torch.equal(t, ref):
                bad[name] += 1
    print(f"permlane={pl}: nondet counts over 15 reruns: {bad}")
