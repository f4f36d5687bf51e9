"""Speculative decoding (draft-assisted generation) — beyond reference parity.

Greedy-exact formulation: a small DRAFT model proposes ``k`` tokens
autoregressively; the TARGET model scores all k proposals in ONE forward
(the eager attention path extends the KV cache by multiple tokens
natively); the longest prefix whose target argmax agrees is accepted, plus
the target's own token at the first disagreement. Caches rewind with
``KVCache.trim``. The output is IDENTICAL to target-only greedy decoding
(tests assert token-for-token equality) — the draft only changes SPEED:
on MI355X, k accepted tokens replace k single-token target forwards with
one (k)-token forward.

Loop invariants (seq = prompt + generated so far):
  * target cache holds KV for exactly ``seq``; ``t_logits`` are the
    target's next-token logits after ``seq``.
  * draft cache holds KV for ``seq[:-len(seed)]`` where ``seed`` (1-2
    tokens) is re-fed at the start of the next proposal pass.

Temperature > 0 would need rejection sampling to keep the target
distribution exact; greedy only for now (raises otherwise — ROADMAP).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..models.llama import Model, make_prompt_cache


@torch.no_grad()
def speculative_generate_tokens(
    model: Model,
    draft_model: Model,
    prompt_tokens: List[int],
    max_tokens: int = 128,
    k: int = 4,
    temperature: float = 0.0,
    stop_tokens: Optional[List[int]] = None,
) -> List[int]:
    """Returns the generated token list (prompt excluded)."""
    if temperature != 0.0:
        raise NotImplementedError("speculative decoding is greedy-exact only "
                                  "(rejection sampling is a ROADMAP item)")
    if k < 1:
        raise ValueError("k must be >= 1")
    device = next(model.parameters()).device
    stop = set(stop_tokens or [])

    cache = make_prompt_cache(model)
    dcache = make_prompt_cache(draft_model)
    y = torch.tensor([prompt_tokens], dtype=torch.long, device=device)

    # prefill: target sees the whole prompt; draft sees all but the seed
    t_logits = model(y, cache=cache)[:, -1, :]
    if y.shape[1] > 1:
        draft_model(y[:, :-1], cache=dcache)
    seed: List[int] = [int(y[0, -1])]

    out: List[int] = []
    while len(out) < max_tokens:
        # 1) draft proposes k tokens (first forward re-feeds the seed)
        proposals: List[int] = []
        d_in = torch.tensor([seed], dtype=torch.long, device=device)
        for _ in range(k):
            d_logits = draft_model(d_in, cache=dcache)[:, -1, :]
            tok = int(d_logits.argmax(dim=-1).item())
            proposals.append(tok)
            d_in = torch.tensor([[tok]], dtype=torch.long, device=device)
        # draft cache grew by len(seed) + (k-1) positions

        # 2) target scores all k proposals in one forward
        prop = torch.tensor([proposals], dtype=torch.long, device=device)
        step_logits = model(prop, cache=cache)[0]  # row i: logits after p1..p(i+1)

        # 3) accept the agreeing prefix + the target's token at divergence
        cur_target = int(t_logits.argmax(dim=-1).item())  # after current seq
        n_acc = 0
        accepted: List[int] = []
        for i, p in enumerate(proposals):
            if p != cur_target:
                break
            accepted.append(p)
            n_acc = i + 1
            cur_target = int(step_logits[i].argmax(dim=-1).item())
        accepted.append(cur_target)

        # 4) rewind. target: keep the accepted proposals, then feed the
        #    divergence token below (restores "cache == seq").
        for c in cache:
            c.trim(k - n_acc)
        #    draft: keep seq_old + p1..p(n_acc-1); the next seed re-feeds the
        #    last two real tokens (or just the divergence token if nothing
        #    was accepted).
        d_trim = (k - n_acc) if n_acc >= 1 else (k - 1)
        for c in dcache:
            c.trim(d_trim)
        seed = accepted[-2:] if n_acc >= 1 else accepted[-1:]

        for tok in accepted:
            out.append(tok)
            if tok in stop or len(out) >= max_tokens:
                return out

        last = torch.tensor([[accepted[-1]]], dtype=torch.long, device=device)
        t_logits = model(last, cache=cache)[:, -1, :]
    return out
