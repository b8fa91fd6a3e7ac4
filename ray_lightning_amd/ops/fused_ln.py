"""FusedLayerNorm + fused residual-add+LayerNorm (CDNA4 row kernels).

Transformer pre-norm blocks compute ``x = x + sublayer(ln(x))``: the
fused op merges the residual add into the next LayerNorm's forward (one
read of each input, one write of x_new and the normalized output) and
the whole chain's backward into two passes (see ops/csrc/fused_ln.hip).
ATen's path is add + vectorized_layer_norm forward and
layer_norm_grad_input + cuComputePartGradGammaBeta (+fold) + add-grad
backward — ~8 ms/step of the GPT-2-XL profile
(profiles/r01_gpt2xl_1gpu_baseline.md).

Falls back to composed torch ops off-GPU / unsupported shapes.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from . import _load_ext


def _ln_fusable(x: torch.Tensor, weight: torch.Tensor) -> bool:
    if not x.is_cuda or x.dtype not in (torch.bfloat16, torch.float32):
        return False
    if weight is None or weight.dtype != torch.float32:
        return False
    v = 8 if x.dtype == torch.bfloat16 else 4
    c = x.shape[-1]
    return c % v == 0 and c // v <= 16384 and _load_ext() is not None


class _FusedLN(torch.autograd.Function):
    """Plain LayerNorm: x -> ln(x)."""

    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = _load_ext()
        x = x.contiguous()
        y, _x_new, mean, invstd = ext.fused_ln_fwd(
            x, None, weight, bias, eps)
        ctx.save_for_backward(x, mean, invstd, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mean, invstd, weight = ctx.saved_tensors
        ext = _load_ext()
        dx, dgamma, dbeta = ext.fused_ln_bwd(
            dy, x, None, weight, mean, invstd)
        return dx, dgamma, dbeta, None


class _FusedAddLN(torch.autograd.Function):
    """(residual, sub_out) -> (x_new = residual + sub_out, ln(x_new))."""

    @staticmethod
    def forward(ctx, residual, sub_out, weight, bias, eps):
        ext = _load_ext()
        y, x_new, mean, invstd = ext.fused_ln_fwd(
            residual.contiguous(), sub_out.contiguous(), weight, bias,
            eps)
        ctx.save_for_backward(x_new, mean, invstd, weight)
        return x_new, y

    @staticmethod
    def backward(ctx, d_xnew, dy):
        x_new, mean, invstd, weight = ctx.saved_tensors
        ext = _load_ext()
        dext = d_xnew.contiguous() if d_xnew is not None else None
        dx, dgamma, dbeta = ext.fused_ln_bwd(
            dy, x_new, dext, weight, mean, invstd)
        # x_new = residual + sub_out: both get the same gradient
        return dx, dx, dgamma, dbeta, None


class FusedLayerNorm(nn.LayerNorm):
    """Drop-in ``nn.LayerNorm`` with the fused HIP path; also provides
    ``forward(sub_out, residual=prev_x)`` -> ``(x_new, ln(x_new))`` for
    pre-norm residual chains."""

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None):
        if _ln_fusable(x, self.weight):
            if residual is None:
                return _FusedLN.apply(x, self.weight, self.bias,
                                      self.eps)
            return _FusedAddLN.apply(residual, x, self.weight,
                                     self.bias, self.eps)
        if residual is None:
            return super().forward(x)
        x_new = residual + x
        return x_new, super().forward(x_new)
