"""KV-cached autoregressive generation.

Parity surface: generate_step / generate_lite / beam_search
(/root/reference/core/generation_lite.py:96-378): chunked prefill
(prefill_step_size), KV cache, samplers + repetition penalty, stop tokens,
TPS/memory stats. Fixes the reference gap where its custom models accept no
``cache`` kwarg — our Model supports cache natively.
"""
from __future__ import annotations

import time
from typing import Callable, Generator, List, Optional, Tuple

import torch

from ..models.llama import Model, make_prompt_cache
from ..ops.sampling import make_logits_processors, make_sampler


@torch.no_grad()
def generate_step(
    model: Model,
    prompt_tokens: List[int],
    max_tokens: int = 128,
    sampler: Optional[Callable] = None,
    logits_processors: Optional[List[Callable]] = None,
    prefill_step_size: int = 512,
    stop_tokens: Optional[List[int]] = None,
    kv_bits: Optional[int] = None,
    kv_group_size: int = 64,
    quantized_kv_start: int = 0,
) -> Generator[int, None, None]:
    from .kv_cache import make_cache, maybe_quantize_kv_cache

    device = next(model.parameters()).device
    sampler = sampler or (lambda logits: logits.reshape(-1).argmax())
    logits_processors = logits_processors or []
    stop_tokens = set(stop_tokens or [])

    cache = make_cache(model, kind="chunked")
    tokens = list(prompt_tokens)
    y = torch.tensor([tokens], dtype=torch.long, device=device)

    # chunked prefill (reference :144-154)
    while y.shape[1] > prefill_step_size:
        model(y[:, :prefill_step_size], cache=cache)
        cache = maybe_quantize_kv_cache(cache, quantized_kv_start, kv_bits, kv_group_size)
        y = y[:, prefill_step_size:]
    logits = model(y, cache=cache)[:, -1, :]

    for _ in range(max_tokens):
        cache = maybe_quantize_kv_cache(cache, quantized_kv_start, kv_bits, kv_group_size)
        for proc in logits_processors:
            logits = proc(tokens, logits)
        tok = int(sampler(logits.reshape(-1)).item())
        tokens.append(tok)
        yield tok
        if tok in stop_tokens:
            return
        y = torch.tensor([[tok]], dtype=torch.long, device=device)
        logits = model(y, cache=cache)[:, -1, :]


@torch.no_grad()
def generate(
    model: Model,
    tokenizer,
    prompt: str,
    max_tokens: int = 128,
    temperature: float = 0.0,
    top_p: float = 1.0,
    min_p: float = 0.0,
    repetition_penalty: Optional[float] = None,
    verbose: bool = False,
) -> Tuple[str, dict]:
    """Returns (generated_text, stats)."""
    was_training = model.training
    model.eval()
    prompt_tokens = [tokenizer.BOS_TOKEN] + tokenizer.tokenize(prompt)
    sampler = make_sampler(temperature, top_p, min_p)
    processors = make_logits_processors(repetition_penalty)
    t0 = time.time()
    out_tokens: List[int] = []
    for tok in generate_step(
        model, prompt_tokens, max_tokens, sampler, processors,
        stop_tokens=[tokenizer.EOS_TOKEN],
    ):
        out_tokens.append(tok)
    dt = time.time() - t0
    if was_training:
        model.train()
    text = tokenizer.detokenize([t for t in out_tokens if t != tokenizer.EOS_TOKEN])
    stats = {
        "prompt_tokens": len(prompt_tokens),
        "generated_tokens": len(out_tokens),
        "tokens_per_second": len(out_tokens) / max(dt, 1e-9),
        "seconds": dt,
    }
    if torch.cuda.is_available():
        stats["peak_memory_gb"] = torch.cuda.max_memory_allocated() / 2**30
    if verbose:
        print(f"[generate] {stats}")
    return text, stats


@torch.no_grad()
def beam_search(
    model: Model,
    tokenizer,
    prompt: str,
    max_tokens: int = 64,
    beam_width: int = 4,
    length_penalty: float = 0.6,
) -> List[Tuple[str, float]]:
    """Batched beam search WITH a KV cache per beam (the reference re-forwards
    the full sequence each step — generation_lite.py:307; fixed here).

    Returns [(text, score)] sorted best-first.
    """
    device = next(model.parameters()).device
    was_training = model.training
    model.eval()
    prompt_tokens = [tokenizer.BOS_TOKEN] + tokenizer.tokenize(prompt)
    eos = tokenizer.EOS_TOKEN

    # beams: (tokens, logprob, cache, finished)
    cache = make_prompt_cache(model)
    y = torch.tensor([prompt_tokens], dtype=torch.long, device=device)
    logits = model(y, cache=cache)[:, -1, :].float().log_softmax(-1)
    topv, topi = logits.reshape(-1).topk(beam_width)
    beams = []
    for v, i in zip(topv.tolist(), topi.tolist()):
        c = make_prompt_cache(model)
        # replay prompt per beam (simple; beams share prompt prefix cost once here)
        model(y, cache=c)
        beams.append(([i], v, c, i == eos))

    for _ in range(max_tokens - 1):
        if all(b[3] for b in beams):
            break
        candidates = []
        for tokens, score, c, finished in beams:
            if finished:
                candidates.append((tokens, score, c, True))
                continue
            step = torch.tensor([[tokens[-1]]], dtype=torch.long, device=device)
            lg = model(step, cache=c)[:, -1, :].float().log_softmax(-1).reshape(-1)
            v, i = lg.topk(beam_width)
            for vv, ii in zip(v.tolist(), i.tolist()):
                candidates.append((tokens + [ii], score + vv, c, ii == eos))
        # dedup by token sequence, keep best
        seen = {}
        for cand in candidates:
            key = tuple(cand[0])
            if key not in seen or cand[1] > seen[key][1]:
                seen[key] = cand
        ranked = sorted(
            seen.values(),
            key=lambda b: b[1] / (len(b[0]) ** length_penalty),
            reverse=True,
        )
        # clone caches for beams that branched from the same parent
        new_beams = []
        used = set()
        for tokens, score, c, finished in ranked[:beam_width]:
            if id(c) in used and not finished:
                nc = make_prompt_cache(model)
                full = torch.tensor([prompt_tokens + tokens], dtype=torch.long, device=device)
                model(full[:, :-1], cache=nc)
                c = nc
            used.add(id(c))
            new_beams.append((tokens, score, c, finished))
        beams = new_beams

    if was_training:
        model.train()
    out = []
    for tokens, score, _c, _f in beams:
        text = tokenizer.detokenize([t for t in tokens if t != eos])
        out.append((text, score / (len(tokens) ** length_penalty)))
    return sorted(out, key=lambda x: x[1], reverse=True)
