"""Isolate flash-attention backward errors per tensor and tile size."""
import math
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_lightning_amd import ops  # noqa: E402

ext = ops._load_ext()


def ref_attn(q, k, v, scale):
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    T = q.shape[2]
    mask = torch.tril(torch.ones(T, T, device=q.device,
                                 dtype=torch.bool))
    s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return p @ v.float(), s, p


def check(B, H, T):
    torch.manual_seed(0)
    hs = 64
    scale = 1.0 / math.sqrt(hs)
    q = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()
    dy = (torch.randn(B, H, T, hs, device="cuda") * 0.5).bfloat16()

    o, lse = ext.flash_attn_fwd(q, k, v, scale)
    dq, dk, dv = ext.flash_attn_bwd(dy, q, k, v, o, lse, scale)

    q2 = q.float().requires_grad_(True)
    k2 = k.float().requires_grad_(True)
    v2 = v.float().requires_grad_(True)
    o2, s2, p2 = ref_attn(q2, k2, v2, scale)
    o2.backward(dy.float())

    # also reference D and dP for intermediate comparison
    D = (dy.float() * o.float()).sum(-1)
    print(f"B{B} H{H} T{T}:")
    for name, got, exp in (("dv", dv, v2.grad), ("dk", dk, k2.grad),
                           ("dq", dq, q2.grad)):
        err = (got.float() - exp).abs()
        rel = err.max() / (exp.abs().max() + 1e-9)
        print(f"  {name}: max_abs={err.max():.4f} "
              f"mean={err.mean():.5f} rel={rel:.4f}")
    # fwd sanity
    ferr = (o.float() - o2.detach()).abs().max()
    print(f"  fwd max err {ferr:.4f}; "
          f"lse err {(lse - torch.logsumexp(s2, -1).detach()).abs().max():.5f}; "
          f"D mean {D.abs().mean():.4f}")


if __name__ == "__main__":
    check(1, 1, 128)
    check(2, 3, 256)
    check(1, 2, 512)
