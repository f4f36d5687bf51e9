import torch

from mlx_cuda_distributed_pretraining_amd.core.config import DataConfig
from mlx_cuda_distributed_pretraining_amd.data import TokenizerManager
from mlx_cuda_distributed_pretraining_amd.inference.generate import beam_search, generate
from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs
from mlx_cuda_distributed_pretraining_amd.ops.sampling import (
    make_logits_processors, sample_token,
)

torch.manual_seed(0)


def setup():
    tok = TokenizerManager(DataConfig())
    args = ModelArgs(hidden_size=32, intermediate_size=64, num_layers=2,
                     num_heads=2, vocab_size=tok.vocab_size)
    return Model(args), tok


def test_generate_produces_text():
    model, tok = setup()
    text, stats = generate(model, tok, "hi", max_tokens=8, temperature=0.0)
    assert isinstance(text, str)
    assert stats["generated_tokens"] <= 8
    assert stats["tokens_per_second"] > 0


def test_generate_greedy_deterministic():
    model, tok = setup()
    t1, _ = generate(model, tok, "abc", max_tokens=6, temperature=0.0)
    t2, _ = generate(model, tok, "abc", max_tokens=6, temperature=0.0)
    assert t1 == t2


def test_generate_sampled():
    model, tok = setup()
    text, _ = generate(model, tok, "abc", max_tokens=6, temperature=1.0, top_p=0.9)
    assert isinstance(text, str)


def test_beam_search_runs():
    model, tok = setup()
    results = beam_search(model, tok, "ab", max_tokens=5, beam_width=2)
    assert len(results) >= 1
    assert results[0][1] >= results[-1][1]  # sorted best-first


def test_sample_token_modes():
    logits = torch.tensor([1.0, 2.0, 10.0, 0.0])
    assert sample_token(logits, temperature=0.0).item() == 2
    torch.manual_seed(0)
    t = sample_token(logits, temperature=1.0, top_p=0.5)
    assert t.item() == 2  # top-p 0.5 keeps only the dominant token
    t = sample_token(logits, temperature=1.0, min_p=0.99)
    assert t.item() == 2


def test_repetition_penalty():
    procs = make_logits_processors(repetition_penalty=2.0, repetition_context_size=4)
    logits = torch.ones(10)
    out = procs[0]([1, 2, 3], logits)
    assert out[1] == 0.5 and out[2] == 0.5 and out[3] == 0.5
    assert out[0] == 1.0


def test_speculative_decoding_matches_greedy():
    """Draft-assisted decoding must be token-for-token identical to
    target-only greedy generation (for ANY draft: the draft only changes
    acceptance rate, never the output). Exercises partial acceptance
    (random draft), full acceptance (draft == target), k=1, and a 1-token
    prompt."""
    import torch
    from mlx_cuda_distributed_pretraining_amd.inference.generate import generate_step
    from mlx_cuda_distributed_pretraining_amd.inference.speculative import (
        speculative_generate_tokens)
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    args = ModelArgs(hidden_size=32, intermediate_size=64, num_layers=2,
                     num_heads=2, num_kv_heads=2, vocab_size=61,
                     max_position_embeddings=128)
    torch.manual_seed(0)
    target = Model(args).eval()
    torch.manual_seed(7)
    draft = Model(args).eval()

    for prompt in ([5, 17, 3, 9], [11]):
        want = list(generate_step(target, prompt, max_tokens=24))
        for d, kk in ((draft, 4), (draft, 1), (draft, 7), (target, 4)):
            got = speculative_generate_tokens(target, d, prompt,
                                              max_tokens=24, k=kk)
            assert got == want, (kk, d is target, got, want)
