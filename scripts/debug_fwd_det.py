"""Localize the fwd/bwd interleave non-determinism: input corruption?"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()
torch.manual_seed(9)
B, H, T = 2, 3, 256
q = torch.randn(B, H, T, 64, device="cuda").bfloat16()
k = torch.randn_like(q)
v = torch.randn_like(q)
dy = torch.randn_like(q)
qc, kc, vc, dyc = q.clone(), k.clone(), v.clone(), dy.clone()

o0, lse0 = ext.flash_attn_fwd_v3(q, k, v, 0.125, True)
oc, lsec = o0.clone(), lse0.clone()
b0 = ext.flash_attn_bwd_v3(dy, q, k, v, o0, lse0, 0.125, True)
b0 = [t.clone() for t in b0]

for i in range(20):
    o, lse = ext.flash_attn_fwd_v3(q, k, v, 0.125, True)
    bb = ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, 0.125, True)
    for name, t, ref in (("q", q, qc), ("k", k, kc), ("v", v, vc),
                         ("dy", dy, dyc), ("o", o, oc),
                         ("lse", lse, lsec)):
        if not torch.equal(t, ref):
            n = (t != ref).sum().item()
            idx = (t != ref).nonzero()[:6].tolist()
            print(f"iter {i}: {name} changed at {n} positions, "
                  f"e.g. {idx}")
            break
    for name, t, ref in zip(("dq", "dk", "dv"), bb, b0):
        if not torch.equal(t, ref):
            n = (t != ref).sum().item()
            idx = (t != ref).nonzero()[:6].tolist()
            print(f"iter {i}: {name} nondet at {n} positions, "
                  f"e.g. {idx}")
            break
print("done")
