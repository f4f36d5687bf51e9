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

Temperature > 0 uses REJECTION SAMPLING (Leviathan et al. / Chen et al.):
the draft SAMPLES its proposals from q_i; proposal i is accepted with
probability min(1, p_i(x)/q_i(x)) and a rejection resamples from the
residual (p_i − q_i)+ — the output is an exact sample from the target
chain regardless of the draft (``_rejection_step`` carries the math and
its own unit tests).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..models.llama import Model, make_prompt_cache


def _rejection_step(p_row: torch.Tensor, q_row: torch.Tensor, tok: int,
                    u: Optional[float] = None):
    """One speculative-sampling acceptance test. Returns None if ``tok``
    (drawn from q) is accepted; otherwise a replacement token sampled from
    the residual (p − q)+ (normalized), which makes the combined law exactly
    p. ``u`` injects the uniform draw for tests."""
    if u is None:
        u = float(torch.rand((), device=p_row.device))
    ratio = float(p_row[tok]) / max(float(q_row[tok]), 1e-20)
    if u <= ratio:
        return None
    resid = (p_row - q_row).clamp_min(0)
    s = float(resid.sum())
    if s <= 0:  # p == q: acceptance was certain; numerical guard
        return None
    return int(torch.multinomial(resid / s, 1).item())


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
        q_rows: List[torch.Tensor] = []
        d_in = torch.tensor([seed], dtype=torch.long, device=device)
        for _ in range(k):
            d_logits = draft_model(d_in, cache=dcache)[:, -1, :]
            if temperature > 0:
                q = torch.softmax(d_logits.float().squeeze(0) / temperature, -1)
                q_rows.append(q)
                tok = int(torch.multinomial(q, 1).item())
            else:
                tok = int(d_logits.argmax(dim=-1).item())
            proposals.append(tok)
            d_in = torch.tensor([[tok]], dtype=torch.long, device=device)
        # draft cache grew by len(seed) + (k-1) positions

        # 2) target scores all k proposals in one forward
        prop = torch.tensor([proposals], dtype=torch.long, device=device)
        step_logits = model(prop, cache=cache)[0]  # row i: logits after p1..p(i+1)

        # 3) accept prefix + divergence token
        n_acc = 0
        accepted: List[int] = []
        if temperature > 0:
            # rejection sampling: p_i is the target dist BEFORE proposal i
            for i, tok in enumerate(proposals):
                row = t_logits.squeeze(0) if i == 0 else step_logits[i - 1]
                p_row = torch.softmax(row.float() / temperature, -1)
                res = _rejection_step(p_row, q_rows[i], tok)
                if res is None:
                    accepted.append(tok)
                    n_acc = i + 1
                else:
                    accepted.append(res)
                    break
            else:
                # every proposal accepted: draw the bonus token from the
                # target's post-prefix distribution
                p_row = torch.softmax(step_logits[k - 1].float() / temperature, -1)
                accepted.append(int(torch.multinomial(p_row, 1).item()))
        else:
            cur_target = int(t_logits.argmax(dim=-1).item())  # after current seq
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
