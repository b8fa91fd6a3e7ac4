"""Time the flash kernels individually on the GPT-2-XL shape."""
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops  # noqa: E402

ext = ops._load_ext()
import os as _os
B = int(_os.environ.get("FB_B", "8"))
H, T, hs = 25, 1024, 64
scale = 1.0 / math.sqrt(hs)
torch.manual_seed(0)
q = torch.randn(B, H, T, hs, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)
dy = torch.randn_like(q)


def t(fn, n=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


o, lse = ext.flash_attn_fwd(q, k, v, scale)
t_fwd = t(lambda: ext.flash_attn_fwd(q, k, v, scale))
t_fwd3s = t(lambda: ext.flash_attn_fwd_v3(q, k, v, scale, False))
t_fwd3p = t(lambda: ext.flash_attn_fwd_v3(q, k, v, scale, True))
t_fwd4 = t(lambda: ext.flash_attn_fwd_v4(q, k, v, scale))
print(f"fwd v3(shfl) {t_fwd3s:.3f} ms   fwd v3(permlane) {t_fwd3p:.3f} ms   fwd v4(32x32) {t_fwd4:.3f} ms")
t_bwd = t(lambda: ext.flash_attn_bwd(dy, q, k, v, o, lse, scale))
t_bwd3s = t(lambda: ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, scale, False))
t_bwd3p = t(lambda: ext.flash_attn_bwd_v3(dy, q, k, v, o, lse, scale, True))
print(f"fwd {t_fwd:.3f} ms   bwd(all3) {t_bwd:.3f} ms")
print(f"bwd v3(shfl) {t_bwd3s:.3f} ms   bwd v3(permlane) {t_bwd3p:.3f} ms")

# SDPA split
qs = q.clone().requires_grad_(True)
ks = k.clone().requires_grad_(True)
vs = v.clone().requires_grad_(True)
t_sf = t(lambda: torch.nn.functional.scaled_dot_product_attention(
    qs, ks, vs, is_causal=True))


def sdpa_full():
    o2 = torch.nn.functional.scaled_dot_product_attention(
        qs, ks, vs, is_causal=True)
    o2.backward(dy)
    qs.grad = ks.grad = vs.grad = None


t_sfull = t(sdpa_full)
print(f"sdpa fwd {t_sf:.3f} ms   sdpa fwd+bwd {t_sfull:.3f} ms "
      f"(bwd ~{t_sfull - t_sf:.3f})")
