"""Enhanced first-order optimizers: AdamW / SGD / Lion.

Algorithm parity with /root/reference/optimizers/enhanced_optimizers.py:
  - decoupled weight decay that SKIPS bias and norm parameters (:88-102),
  - global-norm gradient clipping (:104-119),
  - bias correction + optional AMSGrad (AdamW),
  - optional EMA weight averaging (:67-86),
  - momentum/nesterov (SGD), sign-momentum (Lion).

Implemented as torch.optim.Optimizer subclasses (CPU + GPU correct); the
GPU hot path for large models is the flat fused kernel in
optim/flat_fused.py (csrc/optim.hip), which implements the same math.
"""
from __future__ import annotations

import math
from typing import Iterable, Optional

import torch
from torch.optim import Optimizer


def _is_no_decay(name: str, param: torch.Tensor) -> bool:
    """Bias and norm params (ndim<=1) are excluded from weight decay."""
    return param.ndim <= 1 or name.endswith(".bias") or "norm" in name.lower()


def split_decay_groups(named_params) -> tuple[list, list]:
    decay, no_decay = [], []
    for name, p in named_params:
        if not p.requires_grad:
            continue
        (no_decay if _is_no_decay(name, p) else decay).append(p)
    return decay, no_decay


def global_grad_norm(params: Iterable[torch.Tensor]) -> torch.Tensor:
    device = None
    total = None
    for p in params:
        if p.grad is None:
            continue
        if device is None:
            device = p.grad.device
            total = torch.zeros((), dtype=torch.float32, device=device)
        total += p.grad.float().pow(2).sum()
    if total is None:
        return torch.zeros(())
    return total.sqrt()


def clip_by_global_norm(params: list, max_norm: float) -> Optional[torch.Tensor]:
    if max_norm is None or max_norm <= 0:
        return None
    norm = global_grad_norm(params)
    coef = torch.clamp(max_norm / (norm + 1e-6), max=1.0)
    for p in params:
        if p.grad is not None:
            p.grad.mul_(coef.to(p.grad.dtype))
    return norm


class _EMAMixin:
    def _ema_init(self, ema_decay: float):
        self._ema_decay = ema_decay
        self._ema_params = None

    def _ema_update(self, params):
        if self._ema_decay <= 0:
            return
        if self._ema_params is None:
            self._ema_params = [p.detach().clone().float() for p in params]
        d = self._ema_decay
        for ema, p in zip(self._ema_params, params):
            ema.mul_(d).add_(p.detach().float(), alpha=1 - d)

    def ema_state(self):
        return self._ema_params

    def copy_ema_to(self, params):
        if self._ema_params is None:
            return
        for ema, p in zip(self._ema_params, params):
            p.data.copy_(ema.to(p.dtype))


class AdamWEnhanced(Optimizer, _EMAMixin):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        amsgrad: bool = False,
        max_grad_norm: float = 0.0,
        ema_decay: float = 0.0,
    ):
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, amsgrad=amsgrad
        )
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self._ema_init(ema_decay)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        all_params = [p for g in self.param_groups for p in g["params"]]
        if self.max_grad_norm > 0:
            clip_by_global_norm(all_params, self.max_grad_norm)
        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    if group["amsgrad"]:
                        state["max_exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                state["step"] += 1
                t = state["step"]
                g = p.grad.float()
                m, v = state["exp_avg"], state["exp_avg_sq"]
                m.mul_(b1).add_(g, alpha=1 - b1)
                v.mul_(b2).addcmul_(g, g, value=1 - b2)
                vv = v
                if group["amsgrad"]:
                    torch.maximum(state["max_exp_avg_sq"], v, out=state["max_exp_avg_sq"])
                    vv = state["max_exp_avg_sq"]
                bc1 = 1 - b1**t
                bc2 = 1 - b2**t
                denom = (vv / bc2).sqrt().add_(group["eps"])
                if group["weight_decay"] > 0 and not group.get("no_decay", False):
                    p.mul_(1 - group["lr"] * group["weight_decay"])
                p.add_(((m / bc1) / denom).to(p.dtype), alpha=-group["lr"])
        self._ema_update(all_params)
        return loss


class SGDEnhanced(Optimizer, _EMAMixin):
    def __init__(
        self,
        params,
        lr: float = 1e-2,
        momentum: float = 0.9,
        nesterov: bool = False,
        weight_decay: float = 0.0,
        max_grad_norm: float = 0.0,
        ema_decay: float = 0.0,
    ):
        defaults = dict(lr=lr, momentum=momentum, nesterov=nesterov, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self._ema_init(ema_decay)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        all_params = [p for g in self.param_groups for p in g["params"]]
        if self.max_grad_norm > 0:
            clip_by_global_norm(all_params, self.max_grad_norm)
        for group in self.param_groups:
            mom = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                if group["weight_decay"] > 0 and not group.get("no_decay", False):
                    g = g.add(p.float(), alpha=group["weight_decay"])
                state = self.state[p]
                if mom > 0:
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(p, dtype=torch.float32)
                    buf = state["momentum_buffer"]
                    buf.mul_(mom).add_(g)
                    g = g.add(buf, alpha=mom) if group["nesterov"] else buf
                p.add_(g.to(p.dtype), alpha=-group["lr"])
        self._ema_update(all_params)
        return loss


class LionEnhanced(Optimizer, _EMAMixin):
    def __init__(
        self,
        params,
        lr: float = 1e-4,
        betas=(0.9, 0.99),
        weight_decay: float = 0.0,
        max_grad_norm: float = 0.0,
        ema_decay: float = 0.0,
    ):
        defaults = dict(lr=lr, betas=betas, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.max_grad_norm = max_grad_norm
        self._ema_init(ema_decay)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        all_params = [p for g in self.param_groups for p in g["params"]]
        if self.max_grad_norm > 0:
            clip_by_global_norm(all_params, self.max_grad_norm)
        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if "exp_avg" not in state:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                m = state["exp_avg"]
                update = (b1 * m + (1 - b1) * g).sign_()
                m.mul_(b2).add_(g, alpha=1 - b2)
                if group["weight_decay"] > 0 and not group.get("no_decay", False):
                    p.mul_(1 - group["lr"] * group["weight_decay"])
                p.add_(update.to(p.dtype), alpha=-group["lr"])
        self._ema_update(all_params)
        return loss
