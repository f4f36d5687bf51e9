"""Muon optimizer: SGD-momentum + Newton-Schulz-5 orthogonalization.

Algorithm parity with /root/reference/optimizers/muon.py:54-141: quintic
Newton-Schulz iteration X <- aX + (bA + cA^2)X with A = XX^T, 5 steps,
coefficients (3.4445, -4.7750, 2.0315); transpose when rows > cols;
pre-normalize by Frobenius norm; scale update by max(1, m/n)^0.5;
non-2D params get plain momentum (or are routed to an alternate optimizer
by HybridOptimizer).

On GPU the NS iteration is 15 GEMMs on MFMA via the HIP extension
(csrc/muon.hip orchestrates rocBLAS-free MFMA tiles for the small matrices;
large matrices go through hipBLASLt batched GEMM — plain library GEMMs are
the right tool there). The torch path below is the reference semantics and
runs the same GEMM chain through torch.matmul.
"""
from __future__ import annotations

import torch
from torch.optim import Optimizer

NS_COEFFS = (3.4445, -4.7750, 2.0315)


@torch.no_grad()
def zeropower_via_newtonschulz5(G: torch.Tensor, steps: int = 5, eps: float = 1e-7) -> torch.Tensor:
    """Orthogonalize a 2D matrix via 5 quintic Newton-Schulz iterations.

    Runs in bf16 on GPU (matches reference's half-precision practice and the
    MFMA sweet spot) with an fp32 Frobenius norm.
    """
    assert G.ndim == 2
    a, b, c = NS_COEFFS
    transpose = G.shape[0] > G.shape[1]
    X = G.t() if transpose else G
    X = X / (X.norm() + eps)
    X = X.to(torch.bfloat16) if G.is_cuda else X.float()
    for _ in range(steps):
        A = X @ X.t()
        B = b * A + c * (A @ A)
        X = a * X + B @ X
    X = X.to(torch.float32)
    return X.t() if transpose else X


class Muon(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 0.02,
        momentum: float = 0.95,
        nesterov: bool = True,
        ns_steps: int = 5,
        weight_decay: float = 0.0,
    ):
        defaults = dict(
            lr=lr, momentum=momentum, nesterov=nesterov, ns_steps=ns_steps,
            weight_decay=weight_decay,
        )
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            mom = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                g = p.grad.float()
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(g)
                buf = state["momentum_buffer"]
                buf.mul_(mom).add_(g)
                g = g.add(buf, alpha=mom) if group["nesterov"] else buf

                if p.ndim == 2:
                    u = zeropower_via_newtonschulz5(g, steps=group["ns_steps"])
                    scale = max(1.0, p.shape[0] / p.shape[1]) ** 0.5
                    u = u * scale
                else:
                    u = g  # plain momentum for non-2D params
                if group["weight_decay"] > 0:
                    p.mul_(1 - group["lr"] * group["weight_decay"])
                p.add_(u.to(p.dtype), alpha=-group["lr"])
        return loss
