"""OptimizationManager: build LR schedule + optimizer from the YAML config.

Parity with /root/reference/core/training.py:764-896 — optimizer names:
adamw, adam, sgd, muon, shampoo, hybrid, adamw_enhanced, sgd_enhanced, lion.
"""
from __future__ import annotations

from typing import Any

import torch

from .enhanced import AdamWEnhanced, LionEnhanced, SGDEnhanced, split_decay_groups
from .hybrid import HybridOptimizer
from .muon import Muon
from .schedules import Schedule, build_schedule
from .shampoo import Shampoo, ShampooParams


class OptimizationManager:
    def __init__(self, training_cfg: Any, total_steps: int):
        self.cfg = training_cfg
        self.total_steps = total_steps
        hp = training_cfg.hyperparameters or {}
        self.learning_rate = float(hp.get("learning_rate", 3e-4))
        self.weight_decay = float(hp.get("weight_decay", 0.0))
        self.max_grad_norm = float(hp.get("gradient_clip", hp.get("max_grad_norm", 0.0)) or 0.0)

    def create_scheduler(self) -> Schedule:
        return build_schedule(self.cfg.scheduler, self.learning_rate, self.total_steps)

    def create_optimizer(self, model: torch.nn.Module):
        opt_cfg = self.cfg.optimization or {}
        name = str(opt_cfg.get("optimizer", "adamw")).lower()
        lr = self.learning_rate
        wd = self.weight_decay
        betas = tuple(opt_cfg.get("betas", (0.9, 0.999)))
        eps = float(opt_cfg.get("eps", 1e-8))

        named = list(model.named_parameters())
        decay, no_decay = split_decay_groups(named)
        groups = [
            {"params": decay, "weight_decay": wd},
            {"params": no_decay, "weight_decay": 0.0, "no_decay": True},
        ]

        if name in ("adamw", "adam"):
            return torch.optim.AdamW(
                groups, lr=lr, betas=betas, eps=eps,
                weight_decay=wd if name == "adamw" else 0.0,
            )
        if name == "adamw_enhanced":
            return AdamWEnhanced(
                groups, lr=lr, betas=betas, eps=eps, weight_decay=wd,
                amsgrad=bool(opt_cfg.get("amsgrad", False)),
                max_grad_norm=self.max_grad_norm,
                ema_decay=float(opt_cfg.get("ema_decay", 0.0)),
            )
        if name == "sgd":
            return torch.optim.SGD(
                groups, lr=lr, momentum=float(opt_cfg.get("momentum", 0.9)),
                weight_decay=wd,
            )
        if name == "sgd_enhanced":
            return SGDEnhanced(
                groups, lr=lr, momentum=float(opt_cfg.get("momentum", 0.9)),
                nesterov=bool(opt_cfg.get("nesterov", False)), weight_decay=wd,
                max_grad_norm=self.max_grad_norm,
                ema_decay=float(opt_cfg.get("ema_decay", 0.0)),
            )
        if name == "lion":
            return LionEnhanced(
                groups, lr=lr, betas=tuple(opt_cfg.get("betas", (0.9, 0.99))),
                weight_decay=wd, max_grad_norm=self.max_grad_norm,
                ema_decay=float(opt_cfg.get("ema_decay", 0.0)),
            )
        if name == "muon":
            return HybridOptimizer(
                named,
                matrix_kwargs=dict(
                    lr=lr,
                    momentum=float(opt_cfg.get("momentum", 0.95)),
                    nesterov=bool(opt_cfg.get("nesterov", True)),
                    weight_decay=wd,
                ),
                non_matrix_kwargs=dict(lr=lr * float(opt_cfg.get("adamw_lr_ratio", 0.3)),
                                       betas=betas, eps=eps, weight_decay=wd),
            )
        if name == "shampoo":
            sp = ShampooParams(
                update_period=int(opt_cfg.get("update_period", 10)),
                start_preconditioning_step=int(opt_cfg.get("start_preconditioning_step", 10)),
                max_preconditioner_dim=int(opt_cfg.get("max_preconditioner_dim", 1024)),
                grafting=str(opt_cfg.get("grafting", "adam")),
            )
            return Shampoo(
                groups, lr=lr, momentum=float(opt_cfg.get("momentum", 0.9)),
                weight_decay=wd, hyperparams=sp,
            )
        if name == "hybrid":
            return HybridOptimizer(
                named,
                matrix_kwargs=dict(lr=lr, weight_decay=wd),
                non_matrix_kwargs=dict(lr=lr, betas=betas, eps=eps, weight_decay=wd),
            )
        raise ValueError(f"Unknown optimizer: {name}")
