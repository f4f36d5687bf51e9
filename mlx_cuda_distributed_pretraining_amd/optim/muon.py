"""Muon optimizer: SGD-momentum + Newton-Schulz-5 orthogonalization.

Algorithm parity with /root/reference/optimizers/muon.py:54-141: quintic
Newton-Schulz iteration X <- aX + (bA + cA^2)X with A = XX^T, 5 steps,
coefficients (3.4445, -4.7750, 2.0315); transpose when rows > cols;
pre-normalize by Frobenius norm; scale update by max(1, m/n)^0.5;
non-2D params get plain momentum (or are routed to an alternate optimizer
by HybridOptimizer).

On GPU the NS iteration runs on the hand-written gfx950 MFMA kernels in
csrc/muon.hip (K8): per step three fused launches —
muon_gemm_nt(X,X) for A = X·Xᵀ, muon_gemm_nt(A,A,beta=b,E=A) for the quintic
combine B = b·A + c·A², and muon_gemm_nn_ax for X = a·X + B·X — bf16 storage
with fp32 MFMA accumulation, shapes zero-padded to tile multiples (exact for
the whole chain: padded rows/columns stay zero). The torch composition below
is the reference semantics (and the CPU / fallback path); a GPU numerics
test compares the two.
"""
from __future__ import annotations

import torch
from torch.optim import Optimizer

NS_COEFFS = (3.4445, -4.7750, 2.0315)


def _ns5_hip(X0: torch.Tensor, steps: int, eps: float) -> torch.Tensor:
    """NS-5 chain on the csrc/muon.hip MFMA kernels. X0 is [m,n] with m<=n."""
    from ..ops._ext import require_ext

    ext = require_ext()
    a, b, c = NS_COEFFS
    m, n = X0.shape
    mp = (m + 127) // 128 * 128
    np_ = (n + 127) // 128 * 128
    X = torch.zeros(mp, np_, dtype=torch.bfloat16, device=X0.device)
    X[:m, :n] = (X0 / (X0.norm() + eps)).to(torch.bfloat16)
    X2 = torch.empty_like(X)
    A = torch.empty(mp, mp, dtype=torch.bfloat16, device=X0.device)
    B = torch.empty_like(A)
    none = torch.empty(0, dtype=torch.bfloat16, device=X0.device)
    for _ in range(steps):
        ext.muon_gemm_nt(X, X, A, 1.0, 0.0, none)   # A = X X^T
        ext.muon_gemm_nt(A, A, B, c, b, A)          # B = c*A@A^T + b*A
        ext.muon_gemm_nn_ax(B, X, X2, a)            # X2 = B@X + a*X
        X, X2 = X2, X
    return X[:m, :n].float()


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
    if G.is_cuda:
        from ..ops._ext import require_ext

        if require_ext() is not None:
            X = _ns5_hip(X, steps, eps)
            return X.t() if transpose else X
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
