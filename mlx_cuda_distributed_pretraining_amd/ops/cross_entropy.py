"""Fused cross-entropy over the vocabulary with pad masking.

Replaces the reference's ``cross_entropy + pad-mask + token-count`` Python
composition (/root/reference/core/training.py:1222-1234). The HIP kernel
(csrc/cross_entropy.hip) does one pass per row: online max + sum-exp over
the 32k vocab in fp32, never materializing an fp32 copy of the logits;
backward writes bf16 dlogits = (softmax - onehot) * scale directly.

Returns (loss_mean_over_tokens, ntokens).
"""
from __future__ import annotations

from typing import Tuple

import torch

from ._ext import get_ext, use_hip


def cross_entropy_ref(
    logits: torch.Tensor, targets: torch.Tensor, ignore_index: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    lf = logits.float()
    mask = targets != ignore_index
    ntok = mask.sum()
    safe_targets = targets.masked_fill(~mask, 0)
    lse = torch.logsumexp(lf, dim=-1)
    picked = lf.gather(-1, safe_targets.unsqueeze(-1)).squeeze(-1)
    loss = ((lse - picked) * mask).sum() / ntok.clamp(min=1)
    return loss, ntok


class _FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, targets: torch.Tensor, ignore_index: int):
        n, v = logits.shape
        if use_hip(logits) and logits.shape[-1] % 8 == 0:  # kernel needs V%8==0
            ext = get_ext()
            loss_sum, ntok, lse = ext.ce_fwd(logits.contiguous(), targets.contiguous(), ignore_index)
            ctx.save_for_backward(logits, targets, lse, ntok)
            ctx.ignore_index = ignore_index
            ctx.hip = True
            ntok_f = ntok.clamp(min=1)
            return loss_sum / ntok_f, ntok
        lf = logits.float()
        mask = targets != ignore_index
        ntok = mask.sum()
        safe_targets = targets.masked_fill(~mask, 0)
        lse = torch.logsumexp(lf, dim=-1)
        picked = lf.gather(-1, safe_targets.unsqueeze(-1)).squeeze(-1)
        loss = ((lse - picked) * mask).sum() / ntok.clamp(min=1)
        ctx.save_for_backward(logits, targets, lse, ntok)
        ctx.ignore_index = ignore_index
        ctx.hip = False
        return loss, ntok

    @staticmethod
    def backward(ctx, dloss, _dntok):
        logits, targets, lse, ntok = ctx.saved_tensors
        scale = dloss / ntok.clamp(min=1).to(dloss.dtype)
        if ctx.hip:
            ext = get_ext()
            dlogits = ext.ce_bwd(
                logits.contiguous(), targets.contiguous(), lse, scale, ctx.ignore_index
            )
            return dlogits, None, None
        mask = (targets != ctx.ignore_index).unsqueeze(-1)
        safe_targets = targets.masked_fill(~mask.squeeze(-1), 0)
        p = torch.softmax(logits.float(), dim=-1)
        p.scatter_add_(
            -1,
            safe_targets.unsqueeze(-1),
            -torch.ones_like(safe_targets, dtype=p.dtype).unsqueeze(-1),
        )
        dlogits = (p * mask * scale).to(logits.dtype)
        return dlogits, None, None


def fused_cross_entropy(
    logits: torch.Tensor, targets: torch.Tensor, ignore_index: int = -100
) -> Tuple[torch.Tensor, torch.Tensor]:
    """logits: [N, V] (any leading shape flattened by caller), targets: [N].

    Returns (mean loss over non-ignored tokens, ntokens tensor)."""
    return _FusedCEFn.apply(logits, targets, ignore_index)
