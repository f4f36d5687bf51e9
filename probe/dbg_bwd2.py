import torch, sys
sys.path.insert(0, '.')
from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention

def run(B, Sq, Skv, Hq, Hkv, D, label, **kw):
    torch.manual_seed(0)
    q = torch.randn(B, Sq, Hq, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Skv, Hkv, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Skv, Hkv, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    causal = kw.get("causal", True)
    o = flash_attention(q, k, v, causal=causal, window=kw.get("window"), prefix_len=kw.get("prefix_len"))
    g = torch.randn_like(o)
    o.backward(g)
    gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    o2 = attention_ref(q2, k2, v2, causal=causal, window=kw.get("window"), prefix_len=kw.get("prefix_len"))
    o2.backward(g.float())
    eq, ek, ev = (gq - q2.grad).abs(), (gk - k2.grad).abs(), (gv - v2.grad).abs()
    print(f"{label}: dq={eq.max():.4f} dk={ek.max():.4f} dv={ev.max():.4f}")
    for nm, e in (("dq", eq), ("dk", ek), ("dv", ev)):
        if e.max() > 0.1:
            m = e.amax(dim=(0, 2, 3))  # per seq position
            bad = (m > 0.1).nonzero().flatten().tolist()
            print(f"  {nm} bad rows ({len(bad)}): {bad[:16]}{'...' if len(bad)>16 else ''}")
            am = (e == e.max()).nonzero()[0].tolist()
            print(f"  {nm} argmax at (b,s,h,d)={am}")

run(1,128,128,2,2,128, "a D128 S128")
run(1,256,256,2,2,128, "b D128 S256")
run(2,256,256,4,2,128, "c case1 exact")
run(1,256,256,4,2,64,  "d GQA D64 S256")
run(1,256,256,2,2,64,  "e D64 S256")
run(1,256,256,2,2,64,  "f window", window=64)
run(1,128,128,2,2,64,  "g noncausal", causal=False)
run(1,256,256,2,2,128, "h D128 noncausal", causal=False)
