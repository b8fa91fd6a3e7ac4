"""Chunked linear + cross-entropy for large-vocab LM heads.

The naive GPT-2 loss path materializes the full logits tensor twice
(bf16 GEMM output + ``.float()`` copy: ~1.6 GB at B=8, T=1024,
V=50257) and keeps it alive for backward. This computes the loss in
row chunks — GEMM (hipBLASLt bf16) -> fp32 logsumexp -> loss — keeping
only O(chunk * V) live, and recomputes the chunk logits in backward:

    dlogits = softmax(logits) - onehot(target)
    dx      = dlogits @ W
    dW      = sum_chunks dlogits^T @ x_chunk   (fp32 accumulator)

Numerically identical to ``F.cross_entropy(logits.float(), t)`` with
mean reduction (same bf16 GEMM, same fp32 softmax).
"""
from __future__ import annotations

import torch


class _ChunkedLinearCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, targets, chunk_rows):
        n = x.shape[0]
        loss_sum = x.new_zeros((), dtype=torch.float32)
        for i in range(0, n, chunk_rows):
            xc = x[i:i + chunk_rows]
            logits = (xc @ weight.t()).float()
            lse = torch.logsumexp(logits, dim=-1)
            tgt = logits.gather(
                1, targets[i:i + chunk_rows, None]).squeeze(1)
            loss_sum += (lse - tgt).sum()
        ctx.save_for_backward(x, weight, targets)
        ctx.chunk_rows = chunk_rows
        return loss_sum / n

    @staticmethod
    def backward(ctx, gout):
        x, weight, targets = ctx.saved_tensors
        chunk_rows = ctx.chunk_rows
        n = x.shape[0]
        scale = gout.float() / n
        dx = torch.empty_like(x)
        dw32 = torch.zeros(weight.shape, dtype=torch.float32,
                           device=weight.device)
        for i in range(0, n, chunk_rows):
            xc = x[i:i + chunk_rows]
            logits = (xc @ weight.t()).float()
            probs = torch.softmax(logits, dim=-1)
            probs.scatter_add_(
                1, targets[i:i + chunk_rows, None],
                torch.full((xc.shape[0], 1), -1.0,
                           dtype=torch.float32, device=x.device))
            dlogits = (probs * scale).to(x.dtype)
            dx[i:i + chunk_rows] = dlogits @ weight
            dw32 += (dlogits.t() @ xc).float()
        return dx, dw32.to(weight.dtype), None, None


def chunked_cross_entropy(x: torch.Tensor, weight: torch.Tensor,
                          targets: torch.Tensor,
                          chunk_rows: int = None) -> torch.Tensor:
    """Mean cross-entropy of ``x @ weight.T`` against ``targets``
    without materializing the full logits. ``x``: [N, C] (bf16/f32),
    ``weight``: [V, C], ``targets``: [N] int64."""
    if chunk_rows is None:
        import os
        chunk_rows = int(os.environ.get("RLA_CE_CHUNK_ROWS", "8192"))
    return _ChunkedLinearCE.apply(x, weight, targets, chunk_rows)
