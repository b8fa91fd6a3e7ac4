"""Fused transformer MLP on hipBLASLt epilogues — EXPERIMENTAL, off by
default (``RLA_LT_MLP=1`` opts in).

``fused_mlp(x, W1, b1, W2, b2)`` == ``Linear2(gelu_tanh(Linear1(x)))``
with dGELU and both bias-gradient reductions folded into the GEMM
epilogues (csrc/lt_mlp.hip). Measured verdict on hipBLASLt 1.2/gfx950
(profiles/r02_lt_epilogues.md): this library build's epilogue-fused
NT-wgrad kernels are all tiny-macro-tile (MT32x32) variants — 2.5 ms
vs 0.6 ms per GPT-2-XL wgrad — so the fusion костs far more GEMM time
than the ~7 ms/step of point-wise kernels it removes (59.9k -> 26.7k
tokens/s end to end). Kept as a correct, tested path so the fusion can
be re-evaluated on future hipBLASLt releases.
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

_ext = None
_ext_err = None


def _load():
    global _ext, _ext_err
    if _ext is None and _ext_err is None:
        try:
            import importlib.util
            here = os.path.dirname(os.path.abspath(__file__))
            spec = importlib.util.spec_from_file_location(
                "_lt_mlp", os.path.join(here, "_lt_mlp.so"))
            _ext = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(_ext)
        except Exception as e:  # noqa: BLE001
            _ext_err = str(e)
    return _ext


def lt_mlp_available() -> bool:
    return (torch.cuda.is_available() and _load() is not None
            and os.environ.get("RLA_LT_MLP", "0") == "1")


class _FusedMLP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W1, b1, W2, b2):
        ext = _load()
        y, h, z = ext.lt_mlp_fwd(x, W1, b1, W2, b2)
        ctx.save_for_backward(x, W1, W2, h, z)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, W1, W2, h, z = ctx.saved_tensors
        ext = _load()
        dx, dW1, db1, dW2, db2 = ext.lt_mlp_bwd(dy, x, W1, W2, h, z)
        return dx, dW1, db1, dW2, db2


def fused_mlp(x: torch.Tensor, W1: torch.Tensor, b1: torch.Tensor,
              W2: torch.Tensor, b2: torch.Tensor) -> torch.Tensor:
    """x: [*, C]; W1: [F, C]; W2: [C, F]. GELU is tanh-approx."""
    if (lt_mlp_available() and x.is_cuda
            and x.dtype == torch.bfloat16 and W1.dtype == torch.bfloat16):
        shp = x.shape
        out = _FusedMLP.apply(x.reshape(-1, shp[-1]).contiguous(),
                              W1, b1, W2, b2)
        return out.reshape(*shp[:-1], -1)
    return F.linear(F.gelu(F.linear(x, W1, b1), approximate="tanh"),
                    W2, b2)


class _LtLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, W, b):
        (y,) = _load().lt_linear_fwd(x, W, b)
        ctx.save_for_backward(x, W)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, W = ctx.saved_tensors
        dx, dW, db = _load().lt_linear_bwd(dy, x, W)
        return dx, dW, db


def lt_linear(x: torch.Tensor, W: torch.Tensor,
              b: torch.Tensor) -> torch.Tensor:
    """nn.Linear-equivalent with the bias gradient fused into the wgrad
    GEMM (BGRADB). Falls back to F.linear off the fast path."""
    if (lt_mlp_available() and x.is_cuda and b is not None
            and x.dtype == torch.bfloat16 and W.dtype == torch.bfloat16):
        shp = x.shape
        out = _LtLinear.apply(x.reshape(-1, shp[-1]).contiguous(), W, b)
        return out.reshape(*shp[:-1], -1)
    return F.linear(x, W, b)
