"""Framework-owned compute ops.

GPU path: hand-written CDNA4 HIP kernels (``csrc/``), built in-tree for
gfx950. The HIP extension is REQUIRED on GPU — ops raise rather than
silently falling back to eager PyTorch on a GPU box (set
``RLA_ALLOW_EAGER_GPU=1`` only for A/B debugging). CPU path: plain torch
(used by the gloo test tier).

Kernels (SURVEY.md N3/N6 hand-tuned halves):
- multi-tensor gradient pack (flatten, optional fp32->bf16 cast)
- bucket unpack + 1/world scale (optional bf16->fp32 cast), fused
- fused SGD(momentum) and Adam/AdamW optimizer steps (multi-tensor)
- sharded fused Adam (bf16 params/grads, fp32 master state)
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch

_ext = None
_ext_err: Optional[str] = None


def _load_ext():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from . import _hip_ops  # built in-tree by ops/build.py
        _ext = _hip_ops
    except ImportError as e:
        _ext_err = str(e)
    return _ext


def hip_ext_available() -> bool:
    return _load_ext() is not None


def _require_ext():
    ext = _load_ext()
    if ext is None and os.environ.get("RLA_ALLOW_EAGER_GPU") != "1":
        raise RuntimeError(
            "ray_lightning_amd HIP extension (_hip_ops) is not built but a "
            "GPU op was requested. Build it with "
            "`python -m ray_lightning_amd.ops.build` "
            f"(import error: {_ext_err})")
    return ext


def pack_grads(slots: List[torch.Tensor], grads: List[torch.Tensor]) -> None:
    """Copy each grad (any shape) into its flat bucket slot, casting to
    the comm dtype if needed."""
    if not slots:
        return
    if slots[0].is_cuda:
        ext = _require_ext()
        if ext is not None:
            ext.multi_tensor_pack(slots, [g.reshape(-1) for g in grads])
            return
    flat_grads = [g.reshape(-1) for g in grads]
    if slots[0].dtype == flat_grads[0].dtype:
        torch._foreach_copy_(slots, flat_grads)
    else:
        for s, g in zip(slots, flat_grads):
            s.copy_(g)


def unpack_scale(bucket, scale: float) -> None:
    """Apply the 1/world_size scale to a reduced bucket; when the comm
    dtype differs from the grad dtype, fuse the upcast with the scale."""
    if bucket.flat.is_cuda:
        ext = _require_ext()
        if ext is not None:
            if bucket.flat_grad is not None:
                ext.scale_cast(bucket.flat_grad, bucket.flat, scale)
            else:
                ext.scale_inplace(bucket.flat, scale)
            return
    if bucket.flat_grad is not None:
        bucket.flat_grad.copy_(bucket.flat)
        if scale != 1.0:
            bucket.flat_grad.mul_(scale)
    elif scale != 1.0:
        bucket.flat.mul_(scale)
