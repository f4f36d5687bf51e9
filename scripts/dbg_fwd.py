#!/usr/bin/env python3
"""Diagnose fwd numerics: error structure by (wave row-block, column, tile)."""
import sys, math
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention

dev = "cuda:0"
torch.manual_seed(0)
B, S, H, D = 1, 512, 2, 64
q = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
k = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
v = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
o = flash_attention(q, k, v, causal=True)
ref = attention_ref(q.float(), k.float(), v.float(), causal=True, scale=1.0/math.sqrt(D))
err = (o.float() - ref).abs()   # [B,S,H,D]
print("max err:", err.max().item(), "mean:", err.mean().item())
# error per 32-row block (wave) and per head
e32 = err[0].amax(dim=(1, 2)).reshape(-1, 32).amax(dim=1)
print("per-32-row-block max err:", [f"{x:.3f}" for x in e32.tolist()])
eh = err[0].amax(dim=(0, 2))
print("per-head:", eh.tolist())
ed = err[0].amax(dim=(0, 1)).reshape(-1, 16).amax(dim=1)
print("per-16-dcol:", [f"{x:.3f}" for x in ed.tolist()])
bad = (err[0].amax(dim=2) > 0.05)  # [S,H]
print("bad rows:", bad.any(dim=1).nonzero().flatten().tolist()[:40])
# also single-tile case: S=64 (one kv tile, drain-only path)
for St in (32, 64, 128):
    qq = torch.randn(1, St, 1, 64, device=dev, dtype=torch.bfloat16)
    kk = torch.randn(1, St, 1, 64, device=dev, dtype=torch.bfloat16)
    vv = torch.randn(1, St, 1, 64, device=dev, dtype=torch.bfloat16)
    oo = flash_attention(qq, kk, vv, causal=True)
    rr = attention_ref(qq.float(), kk.float(), vv.float(), causal=True, scale=1.0/8)
    print(f"S={St}: max err {(oo.float()-rr).abs().max().item():.4f}")
