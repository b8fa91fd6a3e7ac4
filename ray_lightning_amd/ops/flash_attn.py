"""Hand-written CDNA4 MFMA flash attention (causal, hs=64, bf16).

``flash_attention(q, k, v)`` is SDPA-shaped: [B, H, T, hs] in/out.
The custom v3 kernels (swapped-operand S^T, in-register permlane
redistribution, ds_read_b64_tr_b16 transposed fragments — see
flash_attn.hip) engage for causal bf16 hs=64 T%64==0 on GPU and are
the DEFAULT: measured 0.421 ms fwd+bwd vs AOTriton SDPA's 0.710 ms on
the GPT-2-XL shape (B=8, H=25, T=1024) — 41% faster, winning both
directions (fwd -12%, bwd -47%); profiles/r02_flash_v3.md.
Anything else falls back to ``F.scaled_dot_product_attention``.
``RLA_FLASH=0`` disables the custom path; ``RLA_FLASH_SHFL=1`` selects
the ds_bpermute redistribution variant (debug).
"""
from __future__ import annotations

import math
import os

import torch
import torch.nn.functional as F

from . import _load_ext


def _usable(q: torch.Tensor) -> bool:
    if os.environ.get("RLA_FLASH", "1") == "0":
        return False
    return (q.is_cuda and q.dtype == torch.bfloat16 and q.dim() == 4
            and q.shape[-1] == 64 and q.shape[2] % 64 == 0
            and _load_ext() is not None
            and hasattr(_load_ext(), "flash_attn_fwd_v3"))


def _use_permlane() -> bool:
    return os.environ.get("RLA_FLASH_SHFL", "0") != "1"


class _FlashAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        ext = _load_ext()
        o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, _use_permlane())
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        ext = _load_ext()
        dq, dk, dv = ext.flash_attn_bwd_v3(dout, q, k, v, o, lse,
                                           ctx.scale, _use_permlane())
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor,
                    v: torch.Tensor) -> torch.Tensor:
    """Causal attention; custom MFMA kernels when usable, SDPA
    otherwise."""
    if _usable(q):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        scale = 1.0 / math.sqrt(q.shape[-1])
        return _FlashAttn.apply(q, k, v, scale)
    return F.scaled_dot_product_attention(q, k, v, is_causal=True)
