"""bwd-v3-only loop for rocprof kernel attribution."""
import math
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()
B, H, T, hs = 8, 25, 1024, 64
scale = 1.0 / math.sqrt(hs)
torch.manual_seed(0)
q = torch.randn(B, H, T, hs, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)
dy = torch.randn_like(q)
o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, True)
torch.cuda.synchronize()
which = os.environ.get("RLA_WHICH", "bwd3")
for _ in range(30):
    if which == "bwd3":
        ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, scale, True)
    elif which == "bwd2":
        ext.flash_attn_bwd(dy, q, k, v, o, lse, scale)
    elif which == "fwd3":
        ext.flash_attn_fwd_v3(q, k, v, scale, True)
    elif which == "fwd4":
        ext.flash_attn_fwd_v4(q, k, v, scale)
torch.cuda.synchronize()
print("done", which)
