"""Llama model family, MI355X-native.

Feature parity with the reference's two model files
(/root/reference/models/llama.py, /root/reference/models/llama_standard.py):
embeddings -> N pre-norm blocks (RMSNorm -> attention -> residual,
RMSNorm -> SwiGLU MLP -> residual) -> final RMSNorm -> tied or separate LM
head with optional logit scaling. Deliberate fixes:
RoPE is ALWAYS applied (the reference's default path builds but never applies
it), and the MLP is standard SwiGLU ``down(silu(gate) * up)``.

Hot-path design for gfx950:
  - BSHD activation layout end-to-end (no transposes),
  - QKV and gate+up projections fused into single GEMMs (hipBLASLt),
  - all non-GEMM ops are HIP kernels (ops/): RoPE, tiled flash attention,
    RMSNorm, SwiGLU epilogue,
  - KV-cached decode supported via ``cache=`` (a gap in the reference, whose
    custom models accept no cache kwarg — SURVEY.md §2.8).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

import torch
import torch.nn as nn

from ..ops.attention import attention_ref, flash_attention, rope_flash_attention_qkv
from ..ops.gemv import FastLinear, linear_fast
from ..ops.rmsnorm import RMSNorm, add_rms_norm
from ..parallel.tp import copy_to_tp, gather_sp, reduce_from_tp, scatter_sp, tp_group
from ..ops.rope import RopeTable, apply_rope
from ..ops.swiglu import swiglu


@dataclass
class ModelArgs:
    hidden_size: int = 512
    intermediate_size: int = 1408
    num_layers: int = 8
    num_heads: int = 8
    num_kv_heads: Optional[int] = None
    head_dim: Optional[int] = None
    max_position_embeddings: Optional[int] = None
    vocab_size: int = 32000
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    rope_traditional: bool = False
    rope_scaling: Optional[float] = None
    attention_bias: bool = False
    mlp_bias: bool = False
    tie_word_embeddings: bool = True
    logit_scale: Optional[float] = None
    attention_type: str = "flash"  # flash | flex | simple
    attention_window: Optional[int] = None  # sliding-window size (flex)
    attention_prefix_len: Optional[int] = None  # prefix-LM split (flex)
    use_alibi: bool = False
    fp8: bool = False  # opt-in e4m3 GEMMs for the block projections (ops/fp8.py)
    router_aux_loss_coef: float = 0.01  # MoE load-balance loss weight
    # MoE knobs exist in the reference config (models/llama.py:40-41) but no
    # MoE layer is implemented there; kept for config parity.
    num_local_experts: int = 0
    num_experts_per_tok: int = 0
    moe_capacity_factor: float = 0.0  # 0 = dropless per-expert loop

    def __post_init__(self):
        if self.num_kv_heads is None:
            self.num_kv_heads = self.num_heads
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_heads

    @classmethod
    def from_config(cls, model_cfg: Any, vocab_size: int) -> "ModelArgs":
        dims: Dict[str, int] = model_cfg.dimensions or {}
        attn: Dict[str, Any] = model_cfg.attention or {}
        norm: Dict[str, float] = model_cfg.normalization or {}
        rope: Dict[str, Any] = model_cfg.rope or {}
        misc: Dict[str, Any] = model_cfg.misc or {}
        return cls(
            hidden_size=int(dims.get("hidden_size", 512)),
            intermediate_size=int(dims.get("intermediate_size", 4 * int(dims.get("hidden_size", 512)))),
            num_layers=int(dims.get("num_layers", 8)),
            num_heads=int(attn.get("num_heads", 8)),
            num_kv_heads=attn.get("num_kv_heads"),
            head_dim=attn.get("head_dim"),
            max_position_embeddings=attn.get("max_position_embeddings"),
            vocab_size=vocab_size,
            rms_norm_eps=float(norm.get("rms_norm_eps", 1e-5)),
            rope_theta=float(rope.get("theta", 10000.0)),
            rope_traditional=bool(rope.get("traditional", False)),
            rope_scaling=rope.get("scaling"),
            attention_bias=bool(misc.get("attention_bias", False)),
            mlp_bias=bool(misc.get("mlp_bias", False)),
            tie_word_embeddings=bool(misc.get("tie_word_embeddings", True)),
            logit_scale=misc.get("logit_scale"),
            fp8=bool(misc.get("fp8", False)),
            router_aux_loss_coef=float(misc.get("router_aux_loss_coef", 0.01)),
            attention_type=str(attn.get("type", "flash")),
            attention_window=attn.get("window"),
            attention_prefix_len=attn.get("prefix_len"),
            use_alibi=bool(attn.get("alibi", False)),
            num_local_experts=int(dims.get("num_local_experts", 0) or 0),
            num_experts_per_tok=int(dims.get("num_experts_per_tok", 0) or 0),
            moe_capacity_factor=float(dims.get("moe_capacity_factor", 0.0) or 0.0),
        )


class KVCache:
    """Simple growing KV cache, BSHD layout ([B, S, Hkv, D])."""

    def __init__(self):
        self.k: Optional[torch.Tensor] = None
        self.v: Optional[torch.Tensor] = None

    @property
    def offset(self) -> int:
        return 0 if self.k is None else self.k.shape[1]

    def trim(self, n: int) -> None:
        """Drop the last n cached positions (speculative-decoding rewind)."""
        if self.k is not None and n > 0:
            self.k = self.k[:, :-n].contiguous()
            self.v = self.v[:, :-n].contiguous()

    def update(self, k: torch.Tensor, v: torch.Tensor):
        if self.k is None:
            self.k, self.v = k, v
        else:
            self.k = torch.cat([self.k, k], dim=1)
            self.v = torch.cat([self.v, v], dim=1)
        return self.k, self.v


def make_prompt_cache(model: "Model") -> List[KVCache]:
    return [KVCache() for _ in range(len(model.layers))]


class Attention(nn.Module):
    def __init__(self, args: ModelArgs, rope_table: RopeTable):
        super().__init__()
        self.args = args
        self.n_heads = args.num_heads
        self.n_kv_heads = args.num_kv_heads
        self.head_dim = args.head_dim
        self.scale = self.head_dim**-0.5
        qkv_out = (self.n_heads + 2 * self.n_kv_heads) * self.head_dim
        self.wqkv = FastLinear(args.hidden_size, qkv_out, bias=args.attention_bias)
        self.wo = FastLinear(self.n_heads * self.head_dim, args.hidden_size, bias=args.attention_bias)
        self.wqkv.fp8 = self.wo.fp8 = args.fp8
        self.rope_table = rope_table
        if args.use_alibi:
            slopes = torch.tensor(
                [2 ** (-8.0 * (i + 1) / self.n_heads) for i in range(self.n_heads)]
            )
            self.register_buffer("alibi_slopes", slopes, persistent=False)
        else:
            self.alibi_slopes = None

    def forward(self, x: torch.Tensor, cache: Optional[KVCache] = None) -> torch.Tensor:
        if getattr(self, "_sp", False) and self.training:
            x = gather_sp(x)  # SP: S-sharded stream -> full sequence
        elif getattr(self, "_tp", False):
            x = copy_to_tp(x)  # f: replicated activation enters column-parallel qkv
        B, S, _ = x.shape
        qkv = self.wqkv(x)

        if getattr(cache, "static_decode", False):
            # hipGraph-capturable decode step (inference/static_decode.py):
            # RoPE at device position + static-cache append + split-KV
            # decode attention, all shape-static
            assert S == 1, "static decode processes one token at a time"
            o = cache.attend(self, qkv)
            return self.wo(o.reshape(B, S, -1))  # static decode is TP-free

        atype = self.args.attention_type
        if cache is None and atype != "simple":
            # training fast path: one fused autograd node over the fused QKV
            # tensor (RoPE + attention; backward writes dQKV in place — no
            # split-backward grad cat)
            cos, sin = self.rope_table.get(S, x.device, 0)
            o = rope_flash_attention_qkv(
                qkv, cos, sin, self.n_heads, self.n_kv_heads, self.head_dim,
                traditional=self.args.rope_traditional, causal=True,
                scale=self.scale,
                window=self.args.attention_window if atype == "flex" else None,
                prefix_len=self.args.attention_prefix_len if atype == "flex" else None,
                alibi_slopes=self.alibi_slopes if self.args.use_alibi else None,
            )
            out = self.wo(o.reshape(B, S, -1))
            if getattr(self, "_sp", False) and self.training:
                return scatter_sp(out)
            return reduce_from_tp(out) if getattr(self, "_tp", False) else out

        q, k, v = qkv.split(
            [
                self.n_heads * self.head_dim,
                self.n_kv_heads * self.head_dim,
                self.n_kv_heads * self.head_dim,
            ],
            dim=-1,
        )
        q = q.view(B, S, self.n_heads, self.head_dim)
        k = k.view(B, S, self.n_kv_heads, self.head_dim)
        v = v.view(B, S, self.n_kv_heads, self.head_dim)

        offset = cache.offset if cache is not None else 0
        cos, sin = self.rope_table.get(S, x.device, offset)
        q = apply_rope(q, cos, sin, self.args.rope_traditional, offset)
        k = apply_rope(k, cos, sin, self.args.rope_traditional, offset)

        if cache is not None:
            k, v = cache.update(k.contiguous(), v.contiguous())

        atype = self.args.attention_type
        if atype == "simple":
            # reference SimpleAttention parity path (fp32 composition)
            o = attention_ref(q, k, v, causal=True, scale=self.scale)
        else:
            o = flash_attention(
                q, k, v,
                causal=True,
                scale=self.scale,
                window=self.args.attention_window if atype == "flex" else None,
                prefix_len=self.args.attention_prefix_len if atype == "flex" else None,
                alibi_slopes=self.alibi_slopes if self.args.use_alibi else None,
            )
        out = self.wo(o.reshape(B, S, -1))
        if getattr(self, "_sp", False) and self.training:
            return scatter_sp(out)
        return reduce_from_tp(out) if getattr(self, "_tp", False) else out


class MLP(nn.Module):
    """Standard SwiGLU with fused gate+up projection GEMM."""

    def __init__(self, args: ModelArgs):
        super().__init__()
        self.w_gate_up = FastLinear(args.hidden_size, 2 * args.intermediate_size, bias=args.mlp_bias)
        self.w_down = FastLinear(args.intermediate_size, args.hidden_size, bias=args.mlp_bias)
        self.w_gate_up.fp8 = self.w_down.fp8 = args.fp8

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        sp = getattr(self, "_sp", False) and self.training
        if sp:
            x = gather_sp(x)
        elif getattr(self, "_tp", False):
            x = copy_to_tp(x)
        # emit_amax: w_down's fp8 quantizer reuses the swiglu-accumulated amax
        out = self.w_down(swiglu(self.w_gate_up(x), emit_amax=self.w_down.fp8))
        if sp:
            return scatter_sp(out)
        return reduce_from_tp(out) if getattr(self, "_tp", False) else out


class MoE(nn.Module):
    """Mixture-of-experts FFN — BEYOND reference parity: the reference carries
    the ``num_local_experts`` / ``num_experts_per_tok`` config knobs but
    implements no MoE layer (/root/reference/models/llama.py:40-41,
    SURVEY.md §2.2).

    Design: per-expert fused gate+up / down weights held as single stacked
    3-D parameters (one flat region each in the fused optimizer), softmax
    top-k routing with renormalized gates, Switch-style load-balancing aux
    loss exposed on ``self.aux_loss`` (the trainer adds
    ``model.aux_loss * router_aux_loss_coef``). The expert loop computes a
    gathered token batch per expert (index_select -> GEMM -> index_add);
    single-GPU 288 GB holds large expert counts without EP — expert-parallel
    all-to-all is the documented round-2 step (ROADMAP.md)."""

    def __init__(self, args: ModelArgs):
        super().__init__()
        E, H, I = args.num_local_experts, args.hidden_size, args.intermediate_size
        self.num_experts = E
        self.top_k = max(1, int(args.num_experts_per_tok or 1))
        if self.top_k > E:
            raise ValueError(f"num_experts_per_tok={self.top_k} > num_local_experts={E}")
        self.router = FastLinear(H, E, bias=False)
        self.w_gate_up = nn.Parameter(torch.empty(E, 2 * I, H))
        self.w_down = nn.Parameter(torch.empty(E, H, I))
        # same depth-scaled init as the dense projections (_init_weights
        # only touches nn.Linear/nn.Embedding, not raw expert Parameters)
        std = 0.02 / math.sqrt(2 * args.num_layers)
        nn.init.normal_(self.w_gate_up, mean=0.0, std=std)
        nn.init.normal_(self.w_down, mean=0.0, std=std)
        self.aux_loss: Optional[torch.Tensor] = None
        self.capacity_factor = float(getattr(args, "moe_capacity_factor", 0.0) or 0.0)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # expert parallelism (parallel/tp.py _shard_experts): activations
        # replicated, experts sharded — this rank computes its local experts'
        # contributions for every token and the g all-reduce sums the
        # partials (no all-to-all needed in this form)
        ep = getattr(self, "_tp", False)
        sp = getattr(self, "_sp", False)
        shard_shape = x.shape
        if sp:
            from ..parallel.tp import gather_sp

            x = gather_sp(x)       # S-shard -> full tokens (exact autograd)
        elif ep:
            x = copy_to_tp(x)
        shape = x.shape
        xf = x.reshape(-1, shape[-1])
        n = xf.shape[0]
        probs = torch.softmax(self.router(xf).float(), dim=-1)  # [N, E] (full E)
        gates, idx = probs.topk(self.top_k, dim=-1)             # [N, k]
        gates = (gates / gates.sum(-1, keepdim=True)).to(x.dtype)
        # Switch aux loss: E * sum_e fraction_routed_e * mean_prob_e  (==1 at
        # perfect balance); computed every forward, consumed by the trainer.
        with torch.no_grad():
            counts = torch.zeros(self.num_experts, device=x.device, dtype=torch.float32)
            counts.scatter_add_(0, idx.reshape(-1),
                                torch.ones(idx.numel(), device=x.device))
            frac = counts / (n * self.top_k)
        self.aux_loss = self.num_experts * (frac * probs.mean(0)).sum()
        if ep:
            # every EP rank computes the FULL aux on the replicated router,
            # and the trainer then SUMS router grads over the group — so the
            # GRADIENT must be pre-divided by the EP degree, while the
            # reported VALUE stays the full aux (detached remainder):
            # value == single-process, grad arrives unscaled after the sum.
            a = self.aux_loss / self._ep_world
            self.aux_loss = a + (self.aux_loss - a).detach()
        n_local = self.w_gate_up.shape[0]
        e0 = self._ep_rank * n_local if ep else 0
        if self.capacity_factor > 0:
            out = self._forward_capacity(xf, gates, idx, e0, n_local)
        else:
            out = torch.zeros_like(xf)
            for el in range(n_local):
                e = e0 + el
                rows, slot = (idx == e).nonzero(as_tuple=True)
                if rows.numel() == 0:
                    continue
                toks = xf.index_select(0, rows)
                h = swiglu(toks @ self.w_gate_up[el].t())
                y = (h @ self.w_down[el].t()) * gates[rows, slot].unsqueeze(-1)
                out.index_add_(0, rows, y.to(out.dtype))
        if sp:
            from ..parallel.tp import scatter_sp

            return scatter_sp(out.reshape(shape)).reshape(shard_shape)
        if ep:
            return reduce_from_tp(out).reshape(shape)
        return out.reshape(shape)

    def _forward_capacity(self, xf: torch.Tensor, gates: torch.Tensor,
                          idx: torch.Tensor, e0: int, n_local: int) -> torch.Tensor:
        """Capacity-bound dispatch (Switch-style): tokens are permuted into
        per-expert slots of fixed capacity C and the expert FFNs run as ONE
        grouped GEMM pair (torch.bmm over [n_local, C, H] on hipBLASLt's
        batched path) instead of a per-expert python loop; tokens beyond an
        expert's capacity are dropped (they keep their other top-k routes).
        Each EP rank dispatches only to its local experts; the surrounding
        reduce_from_tp sums the per-rank partials, so EP stays exact.
        Beyond reference parity (config stubs only at
        /root/reference/models/llama.py:40-41)."""
        n = xf.shape[0]
        C = max(1, int(math.ceil(self.capacity_factor * n * self.top_k / self.num_experts)))
        dev = xf.device
        # flat routing pairs (token t, slot s) -> expert idx[t, s]
        ef = idx.reshape(-1)  # [n*k]
        # position of each pair within its expert's queue (stable arrival order)
        order = torch.argsort(ef, stable=True)
        ranks = torch.empty_like(order)
        ranks[order] = torch.arange(ef.numel(), device=dev) - torch.searchsorted(
            ef[order], torch.arange(self.num_experts, device=dev), side="left"
        ).index_select(0, ef[order])
        local = (ef >= e0) & (ef < e0 + n_local) & (ranks < C)
        pair = local.nonzero(as_tuple=True)[0]           # kept (t,s) pairs
        tok = pair // self.top_k
        slot = pair % self.top_k
        el = ef.index_select(0, pair) - e0               # local expert id
        dst = el * C + ranks.index_select(0, pair)       # slot in the buffer
        H = xf.shape[-1]
        buf = torch.zeros(n_local * C, H, dtype=xf.dtype, device=dev)
        buf.index_copy_(0, dst, xf.index_select(0, tok))
        h = swiglu(torch.bmm(buf.reshape(n_local, C, H),
                             self.w_gate_up.transpose(1, 2)))
        y = torch.bmm(h, self.w_down.transpose(1, 2)).reshape(n_local * C, H)
        g = gates[tok, slot].unsqueeze(-1)
        out = torch.zeros_like(xf)
        out.index_add_(0, tok, (y.index_select(0, dst) * g).to(out.dtype))
        return out


class TransformerBlock(nn.Module):
    def __init__(self, args: ModelArgs, rope_table: RopeTable):
        super().__init__()
        self.attention_norm = RMSNorm(args.hidden_size, args.rms_norm_eps)
        self.attention = Attention(args, rope_table)
        self.mlp_norm = RMSNorm(args.hidden_size, args.rms_norm_eps)
        self.mlp = MoE(args) if args.num_local_experts > 0 else MLP(args)
        # fp8 consumers (wqkv / w_gate_up) reuse the norms' amax accumulators
        self.attention_norm.emit_amax = args.fp8
        self.mlp_norm.emit_amax = args.fp8
        self._checkpoint = False

    def enable_checkpointing(self) -> None:
        """Gradient checkpointing hook the reference calls but its models never
        implement (/root/reference/core/training.py:584-618)."""
        self._checkpoint = True

    def _inner(self, x: torch.Tensor, cache: Optional[KVCache]) -> torch.Tensor:
        x = x + self.attention(self.attention_norm(x), cache)
        x = x + self.mlp(self.mlp_norm(x))
        return x

    def _inner_pair(self, res: torch.Tensor, delta: Optional[torch.Tensor],
                    cache: Optional[KVCache]):
        """Residual-fused path: the incoming stream is (res, delta) with
        x = res + delta; every residual add happens INSIDE the following
        RMSNorm kernel (ops.rmsnorm.add_rms_norm -> csrc/rmsnorm.hip
        HAS_RES), so no standalone elementwise-add kernels run. Returns the
        next (res, delta) pair; the final add folds into the model's last
        norm."""
        if delta is None:
            x = res
            y1 = self.attention_norm(x)
        else:
            x, y1 = add_rms_norm(res, delta, self.attention_norm.weight,
                                 self.attention_norm.eps,
                                 self.attention_norm.emit_amax)
        a = self.attention(y1, cache)
        h, y2 = add_rms_norm(x, a, self.mlp_norm.weight, self.mlp_norm.eps,
                             self.mlp_norm.emit_amax)
        return h, self.mlp(y2)

    def forward_pair(self, res: torch.Tensor, delta: Optional[torch.Tensor],
                     cache: Optional[KVCache] = None):
        if self._checkpoint and self.training and cache is None:
            return torch.utils.checkpoint.checkpoint(
                self._inner_pair, res, delta, cache, use_reentrant=False
            )
        return self._inner_pair(res, delta, cache)

    def _static_decode(self, x: torch.Tensor, cache) -> torch.Tensor:
        """hipGraph decode layer: fused GEMV chain (csrc/gemv.hip gemv_ex) —
        norm1 folds into the QKV GEMV staging, the first residual add into
        the wo epilogue, norm2 into gate-up staging, SwiGLU + the second add
        into the down projection."""
        from ..ops._ext import get_ext

        ext = get_ext()
        attn = self.attention
        none = x.new_empty(0)
        qkv = ext.gemv_ex(x, attn.wqkv.weight, 1, self.attention_norm.weight,
                          self.attention_norm.eps, none)
        o = cache.attend(attn, qkv.unsqueeze(1) if qkv.dim() == 2 else qkv)
        B = x.shape[0]
        o2 = o.reshape(B, -1)
        h = ext.gemv_ex(o2, attn.wo.weight, 0, none, 0.0, x.reshape(B, -1))
        gu = ext.gemv_ex(h, self.mlp.w_gate_up.weight, 1, self.mlp_norm.weight,
                         self.mlp_norm.eps, none)
        out = ext.gemv_ex(gu, self.mlp.w_down.weight, 2, none, 0.0, h)
        return out.view_as(x)

    def forward(self, x: torch.Tensor, cache: Optional[KVCache] = None) -> torch.Tensor:
        if (
            getattr(cache, "static_decode", False)
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and x.shape[1] == 1
            and x.shape[0] <= 8  # batched GEMV limit (LDS staging)
        ):
            return self._static_decode(x[:, 0, :], cache).unsqueeze(1)
        if self._checkpoint and self.training and cache is None:
            return torch.utils.checkpoint.checkpoint(
                self._inner, x, cache, use_reentrant=False
            )
        return self._inner(x, cache)


class Model(nn.Module):
    def __init__(self, args: ModelArgs):
        super().__init__()
        self.args = args
        self.tok_embeddings = nn.Embedding(args.vocab_size, args.hidden_size)
        rope_table = RopeTable(args.head_dim, args.rope_theta, args.rope_scaling)
        self.layers = nn.ModuleList(
            TransformerBlock(args, rope_table) for _ in range(args.num_layers)
        )
        self.norm = RMSNorm(args.hidden_size, args.rms_norm_eps)
        if not args.tie_word_embeddings:
            self.output = FastLinear(args.hidden_size, args.vocab_size, bias=False)
        self.apply(self._init_weights)

    def _init_weights(self, module: nn.Module) -> None:
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, mean=0.0, std=0.02 / math.sqrt(2 * self.args.num_layers))
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, mean=0.0, std=0.02)

    def forward(
        self, tokens: torch.Tensor, cache: Optional[List[KVCache]] = None
    ) -> torch.Tensor:
        vp0 = getattr(self, "_vp_vocab0", -1)
        vp_tied = vp0 >= 0 and self.args.tie_word_embeddings
        if vp_tied:
            from ..parallel.tp import vp_embedding

            x = vp_embedding(tokens, self.tok_embeddings.weight, vp0)
        else:
            x = self.tok_embeddings(tokens)
        static = (
            cache is not None
            and getattr(cache[0], "static_decode", False)
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and x.shape[1] == 1
            and x.shape[0] <= 8
        )
        if not static:
            sp = getattr(self, "_sp_world", 1)
            if sp > 1 and self.training and cache is None:
                # SP entry: keep only this rank's S-rows; the stream stays
                # sharded through norms/residuals (gather/scatter live inside
                # the sublayers). Logits come back SHARDED on S — the trainer
                # computes the shard-local CE and sum-reduces (compute_loss).
                S = x.shape[1]
                if S % sp:
                    raise ValueError(f"sequence_parallel: seq len {S} not divisible by {sp}")
                loc = S // sp
                r = torch.distributed.get_rank(
                    __import__("mlx_cuda_distributed_pretraining_amd.parallel.tp",
                               fromlist=["tp_group"]).tp_group())                     if torch.distributed.is_initialized() else 0
                x = x[:, r * loc:(r + 1) * loc]
            # residual-fused layer chain: adds live inside the RMSNorm kernels
            res, delta = x, None
            for i, layer in enumerate(self.layers):
                res, delta = layer.forward_pair(
                    res, delta, cache[i] if cache is not None else None
                )
            # MoE: surface the summed load-balance loss for the trainer
            aux = [layer.mlp.aux_loss for layer in self.layers
                   if isinstance(layer.mlp, MoE) and layer.mlp.aux_loss is not None]
            self.aux_loss = torch.stack(aux).sum() if aux else None
            if delta is None:
                x = self.norm(res)
            else:
                _, x = add_rms_norm(res, delta, self.norm.weight, self.norm.eps)
            if vp0 >= 0:
                # vocab-parallel lm head (parallel/tp.py): SHARDED logits in
                # training (the trainer runs vocab_parallel_cross_entropy);
                # gathered for eval/generation API compatibility
                x = copy_to_tp(x)
                if self.args.tie_word_embeddings:
                    logits = x @ self.tok_embeddings.weight.t()
                else:
                    logits = self.output(x)
                if not self.training:
                    import torch.distributed as dist_

                    parts = [torch.empty_like(logits) for _ in range(self._tp_world)]
                    dist_.all_gather(parts, logits.contiguous(), group=tp_group())
                    logits = torch.cat(parts, dim=-1)
            elif self.args.tie_word_embeddings:
                logits = linear_fast(x, self.tok_embeddings.weight)
            else:
                logits = self.output(x)
            if self.args.logit_scale:
                logits = logits * self.args.logit_scale
            return logits
        for i, layer in enumerate(self.layers):
            x = layer(x, cache[i] if cache is not None else None)
        if (
            cache is not None
            and getattr(cache[0], "static_decode", False)
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and x.shape[1] == 1
            and x.shape[0] <= 8
        ):
            # decode epilogue: final RMSNorm folded into the lm-head GEMV
            from ..ops._ext import get_ext

            w = (self.tok_embeddings.weight if self.args.tie_word_embeddings
                 else self.output.weight)
            logits = get_ext().gemv_ex(
                x[:, 0, :], w, 1, self.norm.weight, self.norm.eps,
                x.new_empty(0),
            ).view(x.shape[0], 1, -1)
            if self.args.logit_scale:
                logits = logits * self.args.logit_scale
            return logits
        x = self.norm(x)
        if self.args.tie_word_embeddings:
            logits = linear_fast(x, self.tok_embeddings.weight)
        else:
            logits = self.output(x)
        if self.args.logit_scale:
            logits = logits * self.args.logit_scale
        return logits

    @property
    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())

    def load_weights(self, path: str, strict: bool = False) -> None:
        """Non-strict safetensors load (parity:
        /root/reference/models/llama.py:414-477)."""
        from safetensors.torch import load_file

        sd = load_file(path)
        own = self.state_dict()
        filtered = {k: v for k, v in sd.items() if k in own and own[k].shape == v.shape}
        self.load_state_dict(filtered, strict=strict)
