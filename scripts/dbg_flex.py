#!/usr/bin/env python3
import math, sys
from pathlib import Path
import torch
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from mlx_cuda_distributed_pretraining_amd.ops import attention_ref
from mlx_cuda_distributed_pretraining_amd.ops.attention import CompiledBlockMask, flex_attention

dev = "cuda:0"
torch.manual_seed(7)
B, S, H, D = 2, 512, 2, 64

def doc_band_mask(b, h, qi, ki):
    return (ki <= qi) & (qi - ki < 128) & (qi // 200 == ki // 200)

q = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
k = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
v = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
bm = CompiledBlockMask(doc_band_mask, B, H, S, S, device=dev)
print("gran[0,0]:\n", bm.gran[0,0].cpu())
print("range[0,0]:", bm.range[0,0].cpu().tolist())
with torch.no_grad():
    o = flex_attention(q, k, v, block_mask=bm)
ref = attention_ref(q.float(), k.float(), v.float(), causal=False,
                    scale=1.0/math.sqrt(D), mask_mod=doc_band_mask)
err = (o.float()-ref).abs()
print("max err", err.max().item())
e32 = err[0].amax(dim=(1,2)).reshape(-1,32).amax(dim=1)
print("per-32-row:", [f"{x:.3f}" for x in e32.tolist()])
bad = (err[0].amax(dim=(1,2)) > 0.05).nonzero().flatten()
print("bad rows:", bad.tolist()[:30])
