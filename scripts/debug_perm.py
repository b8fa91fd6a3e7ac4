import os
import sys
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops
ext = ops._load_ext()
torch.manual_seed(3)
M = torch.randn(16, 64, device="cuda") * 0.5
Mb = M.bfloat16().float()
for variant in (0, 1):
    out, ident = ext.perm_dump(M, variant)
    err = (out - Mb).abs()
    print(f"variant {variant}: max err {err.max():.4f}, "
          f"bad elems {(err > 0.01).sum().item()}/1024")
    if err.max() > 0.01:
        bad = (err > 0.01).nonzero()[:6].tolist()
        for r, c in bad:
            print(f"   ({r},{c}): got {out[r,c]:.4f} want {Mb[r,c]:.4f}")
print("ident check (expect lane+101... pattern):",
      ident[:8].tolist())
