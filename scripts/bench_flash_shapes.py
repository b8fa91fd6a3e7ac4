"""Flash v3 vs AOTriton SDPA across shapes (generality evidence)."""
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops

ext = ops._load_ext()


def t(fn, n=15):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


print("| B | H | T | v3 fwd | v3 bwd | sdpa fwd | sdpa bwd | total speedup |")
print("|---|---|---|---|---|---|---|---|")
for B, H, T in ((8, 25, 1024), (16, 25, 1024), (8, 25, 512),
                (4, 25, 2048), (2, 12, 4096), (32, 12, 256)):
    hs = 64
    scale = 1.0 / math.sqrt(hs)
    torch.manual_seed(0)
    q = torch.randn(B, H, T, hs, device="cuda", dtype=torch.bfloat16)
    k, v, dy = (torch.randn_like(q) for _ in range(3))
    o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, True)
    f3 = t(lambda: ext.flash_attn_fwd_v3(q, k, v, scale, True))
    b3 = t(lambda: ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, scale,
                                         True))
    qs = q.clone().requires_grad_(True)
    ks = k.clone().requires_grad_(True)
    vs = v.clone().requires_grad_(True)
    sf = t(lambda: torch.nn.functional.scaled_dot_product_attention(
        qs, ks, vs, is_causal=True))

    def sfull():
        out = torch.nn.functional.scaled_dot_product_attention(
            qs, ks, vs, is_causal=True)
        out.backward(dy)
        qs.grad = ks.grad = vs.grad = None

    sfb = t(sfull)
    sb = sfb - sf
    sp = (sf + sb) / (f3 + b3)
    print(f"| {B} | {H} | {T} | {f3:.3f} | {b3:.3f} | {sf:.3f} | "
          f"{sb:.3f} | {sp:.2f}x |")
