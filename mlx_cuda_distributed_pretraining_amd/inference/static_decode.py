"""hipGraph-captured decode: static shapes, device-side position.

The eager decode path launches ~400 kernels per token (22 layers x ~18 ops),
which makes 1B decode launch-bound (~5.3 ms/token measured). Here the WHOLE
decode step — embedding, all layers (RoPE at device position, KV append,
split-KV decode attention, GEMMs), final norm, lm head, greedy argmax, token
ring write, position increment — is captured ONCE into a hipGraph and
replayed per token with zero host work.

Design points (MI355X):
  - the sequence length lives in a device int32 (`pos`); every decode kernel
    reads it from memory, so one captured graph serves all positions,
  - KV caches are preallocated [B, Lmax, Hkv, D] bf16 (288 GB HBM3E: a 1B
    model's full-context cache is ~0.2 GB — capacity is never the issue),
  - attention is a two-phase split-KV flash-decode (csrc/decode.hip):
    chunk partials with own softmax stats, then a combine pass — fills the
    chip even at batch 1,
  - the generated token feeds back through a static input buffer inside the
    graph, and lands in a device ring indexed by pos, so N tokens = N
    graph replays + ONE host sync to read the ring.

CPU (and non-bf16) fall back to an eager composition of the same ops —
used by the unit tests as the numerics oracle.
"""
from __future__ import annotations

from typing import List

import torch

from ..ops._ext import get_ext, use_hip


class StaticLayerCache:
    """Per-layer handle; Attention.forward delegates to ``attend``."""

    def __init__(self, owner: "StaticDecodeState", idx: int):
        self.owner = owner
        self.idx = idx

    # duck-typing marker checked by models.llama.Attention
    static_decode = True

    @property
    def offset(self) -> int:
        return int(self.owner.pos.item())

    def attend(self, attn, qkv: torch.Tensor) -> torch.Tensor:
        """qkv: [B, 1, (Hq+2*Hkv)*D] -> o [B, 1, Hq, D] (post-attention,
        pre-wo). Applies RoPE at the device position, appends k/v to the
        static cache, runs decode attention."""
        ow = self.owner
        B = qkv.shape[0]
        Hq, Hkv, D = attn.n_heads, attn.n_kv_heads, attn.head_dim
        kc, vc = ow.k[self.idx], ow.v[self.idx]
        trad = attn.args.rope_traditional

        if use_hip(qkv, dtypes=(torch.bfloat16,)):
            ext = get_ext()
            qkv = qkv.contiguous()
            qk = qkv[..., : (Hq + Hkv) * D].view(B, 1, Hq + Hkv, D)
            if qk.is_contiguous():  # B == 1: q and k are ADJACENT in the fused
                # projection — one in-place RoPE launch covers both
                ext.rope_decode_(qk, ow.cos, ow.sin, trad, ow.pos)
                q = qkv[..., : Hq * D].view(B, 1, Hq, D)
                k = qkv[..., Hq * D : (Hq + Hkv) * D].view(B, 1, Hkv, D).contiguous()
            else:
                q = qkv[..., : Hq * D].view(B, 1, Hq, D).contiguous()
                k = qkv[..., Hq * D : (Hq + Hkv) * D].view(B, 1, Hkv, D).contiguous()
                ext.rope_decode_(q, ow.cos, ow.sin, trad, ow.pos)
                ext.rope_decode_(k, ow.cos, ow.sin, trad, ow.pos)
            v = qkv[..., (Hq + Hkv) * D :].view(B, 1, Hkv, D).contiguous()
            if ow.kv_bits == 8:
                ksz = ow.ksz[self.idx].view(B, -1)
                vsz = ow.vsz[self.idx].view(B, -1)
                ext.kv_append_q8_(k, v, kc, ksz, vc, vsz, ow.pos)
                return ext.attn_decode_q8(q, kc, ksz, vc, vsz, ow.pos, ow.part,
                                          attn.scale)
            ext.kv_append_(k, v, kc, vc, ow.pos)
            return ext.attn_decode(q, kc, vc, ow.pos, ow.part, attn.scale)

        q = qkv[..., : Hq * D].view(B, 1, Hq, D).contiguous()
        k = qkv[..., Hq * D : (Hq + Hkv) * D].view(B, 1, Hkv, D).contiguous()
        v = qkv[..., (Hq + Hkv) * D :].view(B, 1, Hkv, D).contiguous()

        # eager reference (CPU / non-bf16): same semantics
        from ..ops.attention import attention_ref
        from ..ops.rope import rope_ref

        p = int(ow.pos.item())
        q = rope_ref(q, ow.cos, ow.sin, trad, offset=p)
        k = rope_ref(k, ow.cos, ow.sin, trad, offset=p)
        if ow.kv_bits == 8:
            # mirror of the kv_append_q8/attn_decode_q8 kernels in torch
            def quant_row(t, codes, sz):
                g = t.float().reshape(B, 1, Hkv, D // 64, 64)
                lo = g.amin(-1)
                s = ((g.amax(-1) - lo) / 255.0).clamp_min(1e-8)
                codes[:, p : p + 1] = ((g - lo.unsqueeze(-1)) / s.unsqueeze(-1))                     .round().clamp(0, 255).reshape(B, 1, Hkv, D).to(torch.uint8)
                sz[:, p : p + 1, :, :, 0] = s
                sz[:, p : p + 1, :, :, 1] = lo

            def dequant(codes, sz, n):
                g = codes[:, :n].float().reshape(B, n, Hkv, D // 64, 64)
                s = sz[:, :n, :, :, 0].unsqueeze(-1)
                lo = sz[:, :n, :, :, 1].unsqueeze(-1)
                return (g * s + lo).reshape(B, n, Hkv, D).to(q.dtype)

            quant_row(k, kc, ow.ksz[self.idx])
            quant_row(v, vc, ow.vsz[self.idx])
            kd = dequant(kc, ow.ksz[self.idx], p + 1)
            vd = dequant(vc, ow.vsz[self.idx], p + 1)
            return attention_ref(q, kd, vd, causal=True, scale=attn.scale)
        kc[:, p : p + 1] = k
        vc[:, p : p + 1] = v
        return attention_ref(q, kc[:, : p + 1], vc[:, : p + 1], causal=True,
                             scale=attn.scale)


class StaticDecodeState:
    """Preallocated caches + device position + per-layer handles.

    ``kv_bits=8``: int8 cache with per-64-element-group (scale, zero) —
    dequant FUSED into the decode attention kernel (attn_decode_q8), ~1.8x
    KV memory saving at kernel speed (the eager quantized cache rebuilt the
    full bf16 tensors every step)."""

    def __init__(self, model, batch: int = 1, max_len: int = 2048,
                 kv_bits: int = None):
        args = model.args
        self.model = model
        self.max_len = max_len
        self.kv_bits = kv_bits
        dev = next(model.parameters()).device
        dtype = next(model.parameters()).dtype
        L = len(model.layers)
        Hq, Hkv, D = args.num_heads, args.num_kv_heads, args.head_dim
        if kv_bits == 8:
            assert D % 64 == 0, "int8 KV needs head_dim % 64 == 0"
            self.k = torch.zeros(L, batch, max_len, Hkv, D, device=dev, dtype=torch.uint8)
            self.v = torch.zeros_like(self.k)
            # per-group (scale, zero) as float pairs
            self.ksz = torch.zeros(L, batch, max_len, Hkv, D // 64, 2,
                                   device=dev, dtype=torch.float32)
            self.vsz = torch.zeros_like(self.ksz)
        else:
            assert kv_bits is None, "kv_bits must be None or 8"
            self.k = torch.zeros(L, batch, max_len, Hkv, D, device=dev, dtype=dtype)
            self.v = torch.zeros_like(self.k)
        self.pos = torch.zeros(1, dtype=torch.int32, device=dev)
        nc = (max_len + 255) // 256
        self.part = torch.empty(batch, Hq, nc, D + 2, dtype=torch.float32, device=dev)
        cos, sin = model.layers[0].attention.rope_table.get(max_len, dev, 0)
        self.cos, self.sin = cos.contiguous(), sin.contiguous()
        self.layers: List[StaticLayerCache] = [StaticLayerCache(self, i) for i in range(L)]


class GraphDecoder:
    """Decoder with (on GPU) hipGraph capture of the whole step.

    ``temperature=0`` decodes greedily; otherwise a graph-capturable
    Gumbel-max sampling kernel draws tokens (fresh randomness per replay
    via the device position salt)."""

    def __init__(self, model, batch: int = 1, max_len: int = 2048,
                 temperature: float = 0.0, min_p: float = 0.0, seed: int = 0,
                 kv_bits: int = None):
        if getattr(model.args, "num_local_experts", 0):
            raise NotImplementedError(
                "GraphDecoder: MoE models are not supported by the fused "
                "static-decode path yet (token-dependent routing breaks the "
                "fixed captured graph) — use eager generation"
            )
        if getattr(model, "_tp_world", 1) > 1:
            raise NotImplementedError(
                "GraphDecoder: TP-sharded models need all-reduces inside the "
                "captured GEMV chain (ROADMAP) — decode on a full replica"
            )
        self.model = model
        self.batch = batch
        self.max_len = max_len
        self.temperature = float(temperature)
        self.min_p = float(min_p)
        self.seed = int(seed)
        self.state = StaticDecodeState(model, batch, max_len, kv_bits=kv_bits)
        dev = next(model.parameters()).device
        self.dev = dev
        self.on_gpu = dev.type == "cuda"
        self.in_tok = torch.zeros(batch, 1, dtype=torch.long, device=dev)
        self.ring = torch.zeros(max_len, batch, dtype=torch.long, device=dev)
        self._nxt = torch.zeros(batch, dtype=torch.long, device=dev)
        self.graph = None

    def _pick(self, logits: torch.Tensor) -> torch.Tensor:
        # logits: [B, V]
        if self.temperature <= 0.0:
            return logits.argmax(-1)
        if self.on_gpu and logits.dtype == torch.bfloat16:
            get_ext().sample_token_dev(logits.contiguous(), self._nxt,
                                       self.temperature, self.min_p,
                                       self.seed, self.state.pos)
            return self._nxt
        probs = torch.softmax(logits.float() / self.temperature, -1)
        return torch.multinomial(probs, 1).squeeze(-1)

    @torch.no_grad()
    def prefill(self, prompt_tokens: torch.Tensor) -> None:
        """prompt_tokens: [B, P]. Runs the normal (tiled-attention) path,
        bulk-copies K/V into the static cache, sets pos=P, stages the first
        generated token into the static input buffer."""
        from ..models.llama import make_prompt_cache

        model = self.model
        P = prompt_tokens.shape[1]
        assert P + 1 < self.max_len, "prompt too long for max_len"
        cache = make_prompt_cache(model)
        logits = model(prompt_tokens.to(self.dev), cache=cache)
        for i, c in enumerate(cache):
            if self.state.kv_bits == 8:
                self._bulk_quantize(i, c.k, c.v, P)
            else:
                self.state.k[i][:, :P] = c.k
                self.state.v[i][:, :P] = c.v
        self.state.pos.fill_(P)
        first = self._pick(logits[:, -1])
        self.in_tok.copy_(first.unsqueeze(1))
        self.ring[P % self.max_len] = first

    @torch.no_grad()
    def _bulk_quantize(self, layer: int, k: torch.Tensor, v: torch.Tensor, P: int) -> None:
        """Quantize prompt K/V into the int8 cache (same 64-group scheme as
        the kv_append_q8 kernel)."""
        for t, codes, sz in ((k, self.state.k[layer], self.state.ksz[layer]),
                             (v, self.state.v[layer], self.state.vsz[layer])):
            B, _, H, D = t.shape
            g = t.float().reshape(B, P, H, D // 64, 64)
            lo = g.amin(-1)
            hi = g.amax(-1)
            scale = ((hi - lo) / 255.0).clamp_min(1e-8)
            q = ((g - lo.unsqueeze(-1)) / scale.unsqueeze(-1)).round().clamp(0, 255)
            codes[:, :P] = q.reshape(B, P, H, D).to(torch.uint8)
            sz[:, :P, :, :, 0] = scale
            sz[:, :P, :, :, 1] = lo

    @torch.no_grad()
    def _step(self) -> None:
        logits = self.model(self.in_tok, cache=self.state.layers)
        nxt = self._pick(logits[:, -1])
        # advance pos, then record: the new token lives at ring[pos % cap]
        if self.on_gpu:
            ext = get_ext()
            ext.pos_incr_(self.state.pos)
            ext.write_token_(nxt, self.ring, self.state.pos)
        else:
            self.state.pos += 1
            self.ring[int(self.state.pos.item()) % self.max_len] = nxt
        self.in_tok.copy_(nxt.unsqueeze(1))

    @torch.no_grad()
    def capture(self) -> None:
        """Capture one decode step into a hipGraph (GPU only)."""
        if not self.on_gpu:
            return
        save_pos = self.state.pos.clone()
        save_tok = self.in_tok.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):  # warmup on a side stream
            for _ in range(2):
                self._step()
        torch.cuda.current_stream().wait_stream(s)
        self.state.pos.copy_(save_pos)
        self.in_tok.copy_(save_tok)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._step()
        # capture advanced nothing for real (replay does), but the capture
        # itself ran once logically — reset state again
        self.state.pos.copy_(save_pos)
        self.in_tok.copy_(save_tok)

    @torch.no_grad()
    def decode(self, n_tokens: int) -> torch.Tensor:
        """Generate n_tokens greedily. Returns [B, n_tokens] (device)."""
        p0 = int(self.state.pos.item())
        assert p0 + n_tokens < self.max_len, "decode would exceed max_len"
        if self.graph is not None:
            for _ in range(n_tokens):
                self.graph.replay()
        else:
            for _ in range(n_tokens):
                self._step()
        idx = torch.arange(p0, p0 + n_tokens, device=self.dev) % self.max_len
        out = self.ring[idx]  # [n, B]
        return out.t().contiguous()
