"""FusedBatchNorm2d: hand-written CDNA4 NHWC BN(+ReLU)(+residual).

Replaces the MIOpen BN kernel set plus the separate residual-add and
ReLU elementwise kernels that together dominate the ResNet-50 training
step on MI355X (profiles/r01_resnet50_1gpu_fixedfind.md: BN ~34% +
elementwise ~22% of step time vs convs ~33%). The fused train forward
makes 2 passes over the activation instead of eager's 7 tensor
traversals; the backward 2 instead of 5.

Drop-in: ``FusedBatchNorm2d`` subclasses ``nn.BatchNorm2d`` (same
parameters/buffers/state_dict). The fused HIP path engages on GPU for
channels_last bf16/fp32 inputs; anything else falls back to composed
torch ops so CPU tests and odd shapes stay correct.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from . import _load_ext


def _fusable(x: torch.Tensor, c: int) -> bool:
    if not x.is_cuda or x.dim() != 4:
        return False
    if x.dtype not in (torch.bfloat16, torch.float32):
        return False
    v = 8 if x.dtype == torch.bfloat16 else 4
    if c % v or c // v > 256:
        return False
    return x.is_contiguous(memory_format=torch.channels_last) and \
        _load_ext() is not None


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean,
                running_var, training, momentum, eps, relu):
        ext = _load_ext()
        res_opt = residual
        y, mean, invstd, mask = ext.fused_bn_fwd(
            x, weight, bias,
            running_mean if running_mean is not None else torch.Tensor(),
            running_var if running_var is not None else torch.Tensor(),
            res_opt, relu, training, momentum, eps)
        ctx.relu = relu
        ctx.has_res = residual is not None
        # the ReLU bitmask (1 byte / 8 elems) replaces saving+re-reading
        # y in backward
        ctx.save_for_backward(x, mask, mean, invstd, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, mask, mean, invstd, weight = ctx.saved_tensors
        ext = _load_ext()
        outs = ext.fused_bn_bwd(dy, mask, x, mean, invstd, weight,
                                ctx.relu, ctx.has_res)
        dx, dgamma, dbeta = outs[0], outs[1], outs[2]
        dres = outs[3] if ctx.has_res else None
        return (dx, dres, dgamma, dbeta, None, None, None, None, None,
                None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d with optional fused ReLU and fused residual add.

    ``forward(x)`` == BN(x) [+ReLU]; ``forward(x, residual)`` ==
    ReLU?(BN(x) + residual) — the three fusion shapes a ResNet
    bottleneck needs (bn+relu, plain bn for the downsample branch,
    bn+add+relu for the block output)."""

    def __init__(self, num_features: int, relu: bool = False, **kwargs):
        super().__init__(num_features, **kwargs)
        self.relu = relu

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        training = self.training
        if _fusable(x, self.num_features) and \
                self.weight is not None and \
                self.weight.dtype == torch.float32 and \
                (residual is None or residual.dtype == x.dtype) and \
                (training or (not torch.is_grad_enabled()
                              and self.running_mean is not None)):
            if residual is not None:
                residual = residual.contiguous(
                    memory_format=torch.channels_last)
            if training and self.track_running_stats and \
                    self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            # momentum=None means cumulative moving average
            # (nn.BatchNorm2d semantics): effective factor is
            # 1/num_batches_tracked — the kernel binds a double.
            momentum = self.momentum
            if momentum is None:
                momentum = (1.0 / float(self.num_batches_tracked)
                            if (self.track_running_stats and
                                self.num_batches_tracked is not None and
                                int(self.num_batches_tracked) > 0)
                            else 0.0)
            return _FusedBNFunction.apply(
                x, residual, self.weight, self.bias, self.running_mean,
                self.running_var, training, momentum, self.eps,
                self.relu)
        # composed fallback (CPU, odd channel counts, missing ext)
        out = super().forward(x)
        if residual is not None:
            out = out + residual
        if self.relu:
            out = torch.relu(out)
        return out
