"""Localize fwd v3 non-determinism: where do repeated runs differ?"""
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

for pl in (True, False):
    o0, lse0 = ext.flash_attn_fwd_v3(q, k, v, 0.125, pl)
    diffs = 0
    pos = None
    for i in range(20):
        o, lse = ext.flash_attn_fwd_v3(q, k, v, 0.125, pl)
        if not torch.equal(o, o0):
            diffs += 1
            if pos is None:
                m = (o != o0)
                idx = m.nonzero()
                pos = idx[:20]
        if not torch.equal(lse, lse0):
            print("pl", pl, "iter", i, "LSE differs too")
    print(f"permlane={pl}: {diffs}/20 runs differ")
    if pos is not None:
        print("first mismatch positions (b,h,q,d):")
        for r in pos.tolist():
            print("  ", r, "q%64 =", r[2] % 64, "q%16 =", r[2] % 16,
                  "d%16 =", r[3] % 16)

# v4 determinism too
o0, _ = ext.flash_attn_fwd_v4(q, k, v, 0.125)
d4 = sum(0 if torch.equal(ext.flash_attn_fwd_v4(q, k, v, 0.125)[0], o0)
         else 1 for _ in range(20))
print(f"v4: {d4}/20 runs differ")
# v2 baseline
o0, _ = ext.flash_attn_fwd(q, k, v, 0.125)
d2 = sum(0 if torch.equal(ext.flash_attn_fwd(q, k, v, 0.125)[0], o0)
         else 1 for _ in range(20))
print(f"v2: {d2}/20 runs differ")
