"""Fused optimizer update kernels (flat-buffer, multi-tensor).

Replaces the reference's per-tensor Python tree walks
(/root/reference/optimizers/enhanced_optimizers.py:121-193 AdamW, :302-356 SGD,
:434-487 Lion; grad clip :104-119) with single HIP kernels over FLAT buffers
(csrc/optim.hip): all parameters live as views into one contiguous bf16
tensor, with an fp32 master copy and fp32 moments; the whole update is one
elementwise pass, and global-norm clipping reads the (device-resident)
grad-norm so no host sync happens.

Layout contract: params are ordered [decay params | no-decay params] in the
flat space; callers pass the boundary so weight decay can be applied by
range, not by per-element mask.
"""
from __future__ import annotations

from typing import Optional

import torch

from ._ext import get_ext, use_hip


def grad_sumsq(grad: torch.Tensor) -> torch.Tensor:
    """Sum of squares of the flat grad buffer -> fp32 scalar tensor (device)."""
    if use_hip(grad):
        return get_ext().sumsq(grad)
    return grad.float().pow(2).sum()


def adamw_step(
    param_bf16: torch.Tensor,
    master_f32: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    decay_boundary: int,
    sumsq: Optional[torch.Tensor] = None,
    max_grad_norm: float = 0.0,
) -> None:
    """One fused AdamW step over the flat space (in-place).

    grad may be bf16 or fp32. If sumsq is given and max_grad_norm > 0 the
    kernel rescales grads by min(1, max_norm/sqrt(sumsq)) on the fly.
    """
    if use_hip(param_bf16):
        if sumsq is None:
            sumsq = torch.zeros((), dtype=torch.float32, device=param_bf16.device)
            max_grad_norm = 0.0
        get_ext().adamw_step(
            param_bf16, master_f32, grad, exp_avg, exp_avg_sq,
            sumsq, step, lr, beta1, beta2, eps, weight_decay,
            decay_boundary, max_grad_norm,
        )
        return
    # torch reference path (CPU tests)
    g = grad.float()
    if sumsq is not None and max_grad_norm > 0:
        norm = sumsq.sqrt()
        g = g * torch.clamp(max_grad_norm / (norm + 1e-6), max=1.0)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    upd = (exp_avg / bc1) / denom
    if weight_decay > 0 and decay_boundary > 0:
        master_f32[:decay_boundary].mul_(1 - lr * weight_decay)
    master_f32.add_(upd, alpha=-lr)
    param_bf16.copy_(master_f32.to(param_bf16.dtype))


def lion_step(
    param_bf16: torch.Tensor,
    master_f32: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    weight_decay: float,
    decay_boundary: int,
    sumsq: Optional[torch.Tensor] = None,
    max_grad_norm: float = 0.0,
) -> None:
    """Fused Lion: update = sign(b1*m + (1-b1)*g); m = b2*m + (1-b2)*g."""
    if use_hip(param_bf16):
        if sumsq is None:
            sumsq = torch.zeros((), dtype=torch.float32, device=param_bf16.device)
            max_grad_norm = 0.0
        get_ext().lion_step(
            param_bf16, master_f32, grad, exp_avg, sumsq,
            lr, beta1, beta2, weight_decay, decay_boundary, max_grad_norm,
        )
        return
    g = grad.float()
    if sumsq is not None and max_grad_norm > 0:
        norm = sumsq.sqrt()
        g = g * torch.clamp(max_grad_norm / (norm + 1e-6), max=1.0)
    update = (beta1 * exp_avg + (1 - beta1) * g).sign_()
    exp_avg.mul_(beta2).add_(g, alpha=1 - beta2)
    if weight_decay > 0 and decay_boundary > 0:
        master_f32[:decay_boundary].mul_(1 - lr * weight_decay)
    master_f32.add_(update, alpha=-lr)
    param_bf16.copy_(master_f32.to(param_bf16.dtype))


def sgd_step(
    param_bf16: torch.Tensor,
    master_f32: torch.Tensor,
    grad: torch.Tensor,
    momentum_buf: torch.Tensor,
    step: int,
    lr: float,
    momentum: float,
    weight_decay: float,
    decay_boundary: int,
    nesterov: bool = False,
    sumsq: Optional[torch.Tensor] = None,
    max_grad_norm: float = 0.0,
) -> None:
    if use_hip(param_bf16):
        if sumsq is None:
            sumsq = torch.zeros((), dtype=torch.float32, device=param_bf16.device)
            max_grad_norm = 0.0
        get_ext().sgd_step(
            param_bf16, master_f32, grad, momentum_buf, sumsq,
            lr, momentum, weight_decay, decay_boundary, nesterov, max_grad_norm,
        )
        return
    g = grad.float()
    if sumsq is not None and max_grad_norm > 0:
        norm = sumsq.sqrt()
        g = g * torch.clamp(max_grad_norm / (norm + 1e-6), max=1.0)
    if weight_decay > 0 and decay_boundary > 0:
        g = g.clone()
        g[:decay_boundary] += weight_decay * master_f32[:decay_boundary]
    momentum_buf.mul_(momentum).add_(g)
    upd = g.add(momentum_buf, alpha=momentum) if nesterov else momentum_buf
    master_f32.add_(upd, alpha=-lr)
    param_bf16.copy_(master_f32.to(param_bf16.dtype))
