"""Hybrid optimizer: route 2D matrix params to one optimizer (default Muon),
everything else to another (default AdamW).

Parity with /root/reference/optimizers/hybrid_optimizer.py:60-125, including
optional per-name overrides.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
from torch.optim import Optimizer


class HybridOptimizer(Optimizer):
    def __init__(
        self,
        named_params,
        matrix_optimizer_cls=None,
        non_matrix_optimizer_cls=None,
        matrix_kwargs: Optional[dict] = None,
        non_matrix_kwargs: Optional[dict] = None,
        name_overrides: Optional[Dict[str, str]] = None,  # name -> "matrix"|"non_matrix"
    ):
        from .muon import Muon
        from .enhanced import AdamWEnhanced

        matrix_optimizer_cls = matrix_optimizer_cls or Muon
        non_matrix_optimizer_cls = non_matrix_optimizer_cls or AdamWEnhanced
        named_params = list(named_params)
        overrides = name_overrides or {}
        matrix_params, other_params = [], []
        for name, p in named_params:
            if not p.requires_grad:
                continue
            route = overrides.get(name)
            if route is None:
                route = "matrix" if p.ndim == 2 and "embed" not in name.lower() else "non_matrix"
            (matrix_params if route == "matrix" else other_params).append(p)

        self.matrix_opt = (
            matrix_optimizer_cls(matrix_params, **(matrix_kwargs or {})) if matrix_params else None
        )
        self.non_matrix_opt = (
            non_matrix_optimizer_cls(other_params, **(non_matrix_kwargs or {}))
            if other_params
            else None
        )
        # Register all params so state_dict/zero_grad work through this object.
        super().__init__(
            [p for _, p in named_params if p.requires_grad], defaults={}
        )

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        if self.matrix_opt is not None:
            self.matrix_opt.step()
        if self.non_matrix_opt is not None:
            self.non_matrix_opt.step()
        return loss

    def state_dict(self):
        return {
            "matrix": self.matrix_opt.state_dict() if self.matrix_opt else None,
            "non_matrix": self.non_matrix_opt.state_dict() if self.non_matrix_opt else None,
        }

    def load_state_dict(self, sd):
        if self.matrix_opt is not None and sd.get("matrix") is not None:
            self.matrix_opt.load_state_dict(sd["matrix"])
        if self.non_matrix_opt is not None and sd.get("non_matrix") is not None:
            self.non_matrix_opt.load_state_dict(sd["non_matrix"])
