import torch, sys
sys.path.insert(0, '.')
from mlx_cuda_distributed_pretraining_amd import _mcdp_C as ext

# 1) tr16_frag hardware self-test
for D in (64, 128):
    X = torch.arange(32 * D, device='cuda', dtype=torch.float32).reshape(32, D)
    X = (X % 257 - 128).to(torch.bfloat16)   # asymmetric, exact in bf16
    out = ext.tr16_frag_test(X)              # [2*(D/32), 16, 32]
    ok = True
    Xf = X.float()
    for ks in range(2):
        for dc in range(D // 32):
            frag = out[ks * (D // 32) + dc]  # [16,32]: rows k8, cols c
            want = Xf[ks * 16:ks * 16 + 16, dc * 32:dc * 32 + 32]
            if not torch.equal(frag, want):
                bad = (frag != want).nonzero()[:5]
                print(f"D={D} ks={ks} dc={dc} MISMATCH at {bad.tolist()}")
                print("got ", frag[bad[0][0].item()][:8].tolist())
                print("want", want[bad[0][0].item()][:8].tolist())
                ok = False
    print(f"tr16_frag_test D={D}: {'OK' if ok else 'FAIL'}")

# 2) per-case bwd errors
from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention
cases = [
    (1, 128, 128, 2, 2, 64, {}),
    (2, 256, 256, 4, 2, 128, {}),
    (1, 200, 200, 2, 2, 64, {}),
    (1, 256, 256, 2, 2, 64, {"window": 64}),
    (1, 256, 256, 2, 2, 64, {"prefix_len": 100}),
    (1, 128, 128, 4, 4, 64, {"alibi": True}),
    (1, 1, 96, 2, 2, 64, {}),
    (1, 128, 128, 2, 2, 64, {"causal": False}),
]
for ci, (B, Sq, Skv, Hq, Hkv, D, kw) in enumerate(cases):
    torch.manual_seed(0)
    q = torch.randn(B, Sq, Hq, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Skv, Hkv, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Skv, Hkv, D, device='cuda', dtype=torch.bfloat16, requires_grad=True)
    causal = kw.get("causal", True)
    alibi = Hq if kw.get("alibi") else None
    o = flash_attention(q, k, v, causal=causal, window=kw.get("window"),
                        prefix_len=kw.get("prefix_len"), alibi_slopes=alibi)
    g = torch.randn_like(o)
    o.backward(g)
    gq, gk, gv = q.grad.clone(), k.grad.clone(), v.grad.clone()
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    o2 = attention_ref(q2, k2, v2, causal=causal, window=kw.get("window"),
                       prefix_len=kw.get("prefix_len"), alibi_slopes=alibi)
    o2.backward(g.float())
    errs = [(gq - q2.grad).abs().max().item(), (gk - k2.grad).abs().max().item(),
            (gv - v2.grad).abs().max().item()]
    print(f"case{ci} D={D} kw={kw}: dq={errs[0]:.4f} dk={errs[1]:.4f} dv={errs[2]:.4f}")
