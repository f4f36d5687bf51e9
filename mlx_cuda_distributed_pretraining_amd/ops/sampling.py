"""Sampling ops for generation.

Parity surface: make_sampler / make_logits_processors
(/root/reference/mlx_lm_utils.py:58-145): temperature, top-p, min-p,
repetition penalty. Composed in torch here ([1, V] per decode step is tiny);
the fused single-kernel GPU path (csrc/sampling.hip) covers the hot
temperature+top-p case used by generate().
"""
from __future__ import annotations

from typing import Callable, List, Optional

import torch

from ._ext import get_ext


def sample_token(
    logits: torch.Tensor,
    temperature: float = 0.0,
    top_p: float = 1.0,
    min_p: float = 0.0,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """logits: [V] or [1, V] -> sampled token id (LongTensor [])."""
    logits = logits.reshape(-1).float()
    if temperature <= 0.0:
        return logits.argmax()
    ext = get_ext()
    if logits.is_cuda and ext is not None and hasattr(ext, "sample_token"):
        seed = int(torch.randint(0, 2**31 - 1, (1,), generator=generator).item())
        return ext.sample_token(logits, temperature, top_p, min_p, seed)
    probs = torch.softmax(logits / temperature, dim=-1)
    if min_p > 0.0:
        keep = probs >= min_p * probs.max()
        probs = torch.where(keep, probs, torch.zeros_like(probs))
        probs = probs / probs.sum()
    if top_p < 1.0:
        sorted_probs, sorted_idx = probs.sort(descending=True)
        cum = sorted_probs.cumsum(-1)
        cutoff = (cum - sorted_probs) >= top_p  # drop tokens fully past the mass
        sorted_probs = torch.where(cutoff, torch.zeros_like(sorted_probs), sorted_probs)
        sorted_probs = sorted_probs / sorted_probs.sum()
        pick = torch.multinomial(sorted_probs, 1, generator=generator)
        return sorted_idx[pick].squeeze()
    return torch.multinomial(probs, 1, generator=generator).squeeze()


def make_sampler(
    temp: float = 0.0, top_p: float = 1.0, min_p: float = 0.0,
    generator: Optional[torch.Generator] = None,
) -> Callable[[torch.Tensor], torch.Tensor]:
    def sampler(logits: torch.Tensor) -> torch.Tensor:
        return sample_token(logits, temp, top_p, min_p, generator)

    return sampler


def make_logits_processors(
    repetition_penalty: Optional[float] = None,
    repetition_context_size: int = 20,
) -> List[Callable]:
    processors: List[Callable] = []
    if repetition_penalty is not None and repetition_penalty != 1.0:

        def rep_penalty(tokens: List[int], logits: torch.Tensor) -> torch.Tensor:
            if not tokens:
                return logits
            ctx = torch.tensor(
                tokens[-repetition_context_size:], device=logits.device, dtype=torch.long
            )
            flat = logits.reshape(-1)
            picked = flat[ctx]
            flat = flat.clone()
            flat[ctx] = torch.where(
                picked > 0, picked / repetition_penalty, picked * repetition_penalty
            )
            return flat.reshape(logits.shape)

        processors.append(rep_penalty)
    return processors
