"""Shampoo: Kronecker-factored second-order preconditioning.

Algorithm parity with /root/reference/optimizers/shampoo.py:
  - EMA of GG^T (left) and G^TG (right) statistics (:229-255),
  - inverse-4th-root preconditioners via eigendecomposition-free coupled
    Newton iteration (:88-126),
  - periodic recompute (update_period) after start_preconditioning_step
    (:210-227),
  - preconditioned update L @ G @ R (:257-295),
  - norm grafting onto adam / sgd / momentum (:297-312),
  - decoupled weight decay, max_preconditioner_dim cap (dims above the cap
    fall back to diagonal/grafted update).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
from torch.optim import Optimizer


@dataclass
class ShampooParams:
    beta2: float = 0.99
    epsilon: float = 1e-12
    update_period: int = 10
    start_preconditioning_step: int = 10
    max_preconditioner_dim: int = 1024
    grafting: str = "adam"  # adam | sgd | momentum | none
    graft_beta1: float = 0.9
    graft_beta2: float = 0.999
    graft_eps: float = 1e-8
    exponent_multiplier: float = 1.0


@torch.no_grad()
def matrix_inverse_pth_root(
    A: torch.Tensor, p: int = 4, eps: float = 1e-12, iters: int = 20, ridge: float = 1e-6
) -> torch.Tensor:
    """A^(-1/p) for symmetric PSD A by coupled Newton iteration (fp32/fp64-free).

    X_{k+1} = X_k ((p+1)I - M_k)/p,  M_{k+1} = ((p+1)I - M_k / p)^p M_k
    with A normalized by its trace-based spectral bound for convergence.
    """
    n = A.shape[0]
    I = torch.eye(n, dtype=torch.float32, device=A.device)
    A = A.float() + ridge * I * A.diagonal().mean().clamp(min=eps)
    # normalize: z in (0, 1/||A||]
    norm = A.abs().sum(dim=1).max()  # inf-norm bound on spectral radius
    z = 1.0 / norm.clamp(min=eps)
    X = (z ** (1.0 / p)) * I
    M = z * A
    for _ in range(iters):
        T = ((p + 1) * I - M) / p
        X = X @ T
        M = torch.linalg.matrix_power(T, p) @ M
        if (M - I).abs().max() < 1e-6:
            break
    return X


def _stats_update_hip(S: torch.Tensor, G: torch.Tensor, beta2: float, left: bool) -> None:
    """K9: S = beta2*S + (1-beta2) * X @ X^T on the hand-written gfx950 MFMA
    NT kernel (csrc/muon.hip, fp32-state instantiation), X = G or G^T.

    Inputs are bf16-rounded for the matrix cores (fp32 accumulation, fp32
    EMA state — the ~0.4% input rounding is far below the stats EMA noise);
    shapes are zero-padded to tile multiples (exact: padded rows stay zero).
    Parity: /root/reference/optimizers/shampoo.py:229-255.
    """
    from ..ops._ext import require_ext

    ext = require_ext()
    X = G if left else G.t()
    d, k = X.shape
    dp = (d + 127) // 128 * 128
    kp = (k + 63) // 64 * 64
    Gp = torch.zeros(dp, kp, dtype=torch.bfloat16, device=G.device)
    Gp[:d, :k] = X
    if dp == d:
        ext.shampoo_stats_update(Gp, S, beta2)
    else:
        Sp = torch.zeros(dp, dp, dtype=torch.float32, device=G.device)
        Sp[:d, :d] = S
        ext.shampoo_stats_update(Gp, Sp, beta2)
        S.copy_(Sp[:d, :d])


class Shampoo(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        momentum: float = 0.9,
        weight_decay: float = 0.0,
        hyperparams: Optional[ShampooParams] = None,
    ):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.hp = hyperparams or ShampooParams()

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        hp = self.hp
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["momentum_buffer"] = torch.zeros_like(g)
                    if hp.grafting == "adam":
                        state["graft_m"] = torch.zeros_like(g)
                        state["graft_v"] = torch.zeros_like(g)
                    elif hp.grafting in ("sgd", "momentum"):
                        state["graft_m"] = torch.zeros_like(g)
                    if g.ndim == 2 and max(g.shape) <= hp.max_preconditioner_dim:
                        m, n = g.shape
                        state["stat_l"] = torch.zeros(m, m, dtype=torch.float32, device=g.device)
                        state["stat_r"] = torch.zeros(n, n, dtype=torch.float32, device=g.device)
                        state["prec_l"] = torch.eye(m, dtype=torch.float32, device=g.device)
                        state["prec_r"] = torch.eye(n, dtype=torch.float32, device=g.device)
                state["step"] += 1
                t = state["step"]

                # grafting direction (provides the update NORM)
                if hp.grafting == "adam":
                    gm, gv = state["graft_m"], state["graft_v"]
                    gm.mul_(hp.graft_beta1).add_(g, alpha=1 - hp.graft_beta1)
                    gv.mul_(hp.graft_beta2).addcmul_(g, g, value=1 - hp.graft_beta2)
                    bc1 = 1 - hp.graft_beta1**t
                    bc2 = 1 - hp.graft_beta2**t
                    graft_dir = (gm / bc1) / ((gv / bc2).sqrt() + hp.graft_eps)
                elif hp.grafting in ("sgd", "momentum"):
                    gm = state["graft_m"]
                    gm.mul_(group["momentum"]).add_(g)
                    graft_dir = gm
                else:
                    graft_dir = g

                use_prec = "stat_l" in state
                if use_prec:
                    # statistics EMA (K9 MFMA syrk kernel on GPU)
                    if g.is_cuda:
                        _stats_update_hip(state["stat_l"], g, hp.beta2, left=True)
                        _stats_update_hip(state["stat_r"], g, hp.beta2, left=False)
                    else:
                        state["stat_l"].mul_(hp.beta2).add_(g @ g.t(), alpha=1 - hp.beta2)
                        state["stat_r"].mul_(hp.beta2).add_(g.t() @ g, alpha=1 - hp.beta2)
                    if t >= hp.start_preconditioning_step and (
                        t % hp.update_period == 0 or t == hp.start_preconditioning_step
                    ):
                        state["prec_l"] = matrix_inverse_pth_root(state["stat_l"], p=4, eps=hp.epsilon)
                        state["prec_r"] = matrix_inverse_pth_root(state["stat_r"], p=4, eps=hp.epsilon)

                if use_prec and t >= hp.start_preconditioning_step:
                    update = state["prec_l"] @ g @ state["prec_r"]
                    # norm grafting
                    gnorm = graft_dir.norm()
                    unorm = update.norm().clamp(min=1e-12)
                    update = update * (gnorm / unorm)
                else:
                    update = graft_dir

                buf = state["momentum_buffer"]
                buf.mul_(group["momentum"]).add_(update)
                if group["weight_decay"] > 0:
                    p.mul_(1 - group["lr"] * group["weight_decay"])
                p.add_(buf.to(p.dtype), alpha=-group["lr"])
        return loss
