"""Fused optimizers backed by the CDNA4 multi-tensor kernels.

On GPU these run one (or a few) kernel launches per step over all
parameters — HBM-bandwidth-bound, 16B/lane vectorized — instead of
per-parameter eager op chains. On CPU they fall back to torch foreach
math so the gloo test tier runs everywhere.

``ShardedFusedAdam`` is the sharded-path optimizer:
bf16 model params + fp32 master copy + fp32 moments, one fused kernel.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
from torch.optim import Optimizer

from . import ops


def _use_ext(params: List[torch.Tensor]) -> bool:
    return bool(params) and params[0].is_cuda and ops.hip_ext_available()


class FusedSGD(Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            params, grads, momenta = [], [], []
            first_step = False
            for p in group["params"]:
                if p.grad is None:
                    continue
                params.append(p)
                grads.append(p.grad.reshape(-1))
                if group["momentum"] != 0:
                    state = self.state[p]
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(
                            p, memory_format=torch.contiguous_format)
                        first_step = True
                    momenta.append(
                        state["momentum_buffer"].reshape(-1))
            if not params:
                continue
            flat_params = [p.reshape(-1) for p in params]
            if _use_ext(flat_params) and params[0].dtype == torch.float32:
                ops._load_ext().fused_sgd(
                    flat_params, grads, momenta, group["lr"],
                    group["momentum"], group["dampening"],
                    group["weight_decay"], group["nesterov"], first_step)
            else:
                self._eager_step(flat_params, grads, momenta, group,
                                 first_step)
        return loss

    @staticmethod
    def _eager_step(params, grads, momenta, group, first_step):
        wd, mu = group["weight_decay"], group["momentum"]
        damp, lr = group["dampening"], group["lr"]
        if wd != 0:
            grads = torch._foreach_add(grads, params, alpha=wd)
        if mu != 0:
            if first_step:
                for m, g in zip(momenta, grads):
                    m.copy_(g)
            else:
                torch._foreach_mul_(momenta, mu)
                torch._foreach_add_(momenta, grads, alpha=1 - damp)
            if group["nesterov"]:
                grads = torch._foreach_add(grads, momenta, alpha=mu)
            else:
                grads = momenta
        torch._foreach_add_(params, grads, alpha=-lr)


class FusedAdamW(Optimizer):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 1e-2,
                 adamw: bool = True):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            params, grads, avgs, sqs = [], [], [], []
            step_t = 0
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "exp_avg" not in state:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(
                        p, memory_format=torch.contiguous_format)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, memory_format=torch.contiguous_format)
                state["step"] += 1
                step_t = state["step"]
                params.append(p.reshape(-1))
                grads.append(p.grad.reshape(-1))
                avgs.append(state["exp_avg"].reshape(-1))
                sqs.append(state["exp_avg_sq"].reshape(-1))
            if not params:
                continue
            b1, b2 = group["betas"]
            if _use_ext(params) and params[0].dtype == torch.float32:
                ops._load_ext().fused_adam(
                    params, grads, avgs, sqs, group["lr"], b1, b2,
                    group["eps"], group["weight_decay"], step_t,
                    group["adamw"])
            else:
                self._eager_step(params, grads, avgs, sqs, group, step_t)
        return loss

    @staticmethod
    def _eager_step(params, grads, avgs, sqs, group, step):
        b1, b2 = group["betas"]
        lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
        bc1 = 1 - b1 ** step
        bc2 = 1 - b2 ** step
        if group["adamw"]:
            torch._foreach_mul_(params, 1 - lr * wd)
        elif wd != 0:
            grads = torch._foreach_add(grads, params, alpha=wd)
        torch._foreach_mul_(avgs, b1)
        torch._foreach_add_(avgs, grads, alpha=1 - b1)
        torch._foreach_mul_(sqs, b2)
        torch._foreach_addcmul_(sqs, grads, grads, value=1 - b2)
        denom = torch._foreach_sqrt(sqs)
        torch._foreach_div_(denom, math.sqrt(bc2))
        torch._foreach_add_(denom, eps)
        steps = torch._foreach_div(avgs, denom)
        torch._foreach_add_(params, steps, alpha=-lr / bc1)


class ShardedFusedAdam(Optimizer):
    """Adam(W) for bf16 models: bf16 params are mirrored by an fp32
    master copy; the fused kernel updates master state and writes back
    bf16 — the optimizer the sharded strategy pairs with the GPT-2-XL
    config (BASELINE.json config 4)."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 1e-2,
                 adamw: bool = True):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, adamw=adamw)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            p16, masters, grads, avgs, sqs = [], [], [], [], []
            f32_params, f32_grads, f32_avgs, f32_sqs = [], [], [], []
            step_t = 0
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "exp_avg" not in state:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32,
                        memory_format=torch.contiguous_format)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32,
                        memory_format=torch.contiguous_format)
                    if p.dtype == torch.bfloat16:
                        state["master"] = p.detach().float().clone()
                state["step"] += 1
                step_t = state["step"]
                if p.dtype == torch.bfloat16:
                    p16.append(p.reshape(-1))
                    masters.append(state["master"].reshape(-1))
                    grads.append(p.grad.reshape(-1))
                    avgs.append(state["exp_avg"].reshape(-1))
                    sqs.append(state["exp_avg_sq"].reshape(-1))
                else:
                    f32_params.append(p.reshape(-1))
                    f32_grads.append(p.grad.reshape(-1))
                    f32_avgs.append(state["exp_avg"].reshape(-1))
                    f32_sqs.append(state["exp_avg_sq"].reshape(-1))
            b1, b2 = group["betas"]
            if p16:
                if _use_ext(p16):
                    ops._load_ext().sharded_adam(
                        p16, masters, grads, avgs, sqs, group["lr"], b1,
                        b2, group["eps"], group["weight_decay"], step_t,
                        group["adamw"])
                else:
                    self._eager_bf16(p16, masters, grads, avgs, sqs,
                                     group, step_t)
            if f32_params:
                if _use_ext(f32_params):
                    ops._load_ext().fused_adam(
                        f32_params, f32_grads, f32_avgs, f32_sqs,
                        group["lr"], b1, b2, group["eps"],
                        group["weight_decay"], step_t, group["adamw"])
                else:
                    FusedAdamW._eager_step(f32_params, f32_grads,
                                           f32_avgs, f32_sqs, group,
                                           step_t)
        return loss

    @staticmethod
    def _eager_bf16(p16, masters, grads, avgs, sqs, group, step):
        b1, b2 = group["betas"]
        lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]
        bc1 = 1 - b1 ** step
        bc2 = 1 - b2 ** step
        for p, mst, g, m, v in zip(p16, masters, grads, avgs, sqs):
            gf = g.float()
            if group["adamw"]:
                mst.mul_(1 - lr * wd)
            else:
                gf = gf + wd * mst
            m.mul_(b1).add_(gf, alpha=1 - b1)
            v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
            denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
            mst.addcdiv_(m, denom, value=-lr / bc1)
            p.copy_(mst.to(torch.bfloat16))
