#!/usr/bin/env python3
"""Diagnose muon_gemm_nt / nn_ax numerics: error magnitude + structure."""
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from mlx_cuda_distributed_pretraining_amd.ops._ext import require_ext  # noqa: E402

ext = require_ext()
dev = "cuda:0"
torch.manual_seed(0)

M = N = 128
K = 64
X = (torch.randn(M, K, device=dev) / K**0.5).to(torch.bfloat16)
Y = (torch.randn(N, K, device=dev) / K**0.5).to(torch.bfloat16)
C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
none = torch.empty(0, dtype=torch.bfloat16, device=dev)
ext.muon_gemm_nt(X, Y, C, 1.0, 0.0, none)
ref = X.float() @ Y.float().t()
err = (C.float() - ref).abs()
print("max err:", err.max().item(), " mean:", err.mean().item())
print("err vs ref^T:", (C.float() - ref.t()).abs().max().item())
# per-32x32-quadrant error map (waves/frags)
em = err.reshape(4, 32, 4, 32).amax(dim=(1, 3))
print("per-32-block err max:\n", em)
# column/row pattern of worst errors
bad = (err > 0.05)
print("bad count:", bad.sum().item(), "of", err.numel())
if bad.any():
    idx = bad.nonzero()[:8]
    for i, j in idx:
        print(f"  C[{i},{j}]={C[i,j].item():.4f} ref={ref[i,j].item():.4f}")

# nn_ax
B = (torch.randn(M, M, device=dev) / M**0.5).to(torch.bfloat16)
X2 = torch.randn(M, N, device=dev).to(torch.bfloat16)
C2 = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
ext.muon_gemm_nn_ax(B, X2, C2, 3.4445)
ref2 = B.float() @ X2.float() + 3.4445 * X2.float()
e2 = (C2.float() - ref2).abs()
print("nn_ax max err:", e2.max().item(), "vs refT:",
      (C2.float() - (B.float().t() @ X2.float() + 3.4445 * X2.float())).abs().max().item())
em2 = e2.reshape(4, 32, 4, 32).amax(dim=(1, 3))
print("nn_ax per-32-block err max:\n", em2)
