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


def test_rejection_step_law():
    """Speculative-sampling math (inference/speculative.py _rejection_step):
    the combined law (draft proposal -> accept/residual-resample) must be
    exactly the target distribution p, for an arbitrary draft q."""
    from mlx_cuda_distributed_pretraining_amd.inference.speculative import (
        _rejection_step)

    torch.manual_seed(0)
    V = 8
    p = torch.rand(V); p = p / p.sum()
    q = torch.rand(V) ** 2; q = q / q.sum()
    # p == q: always accepted
    for tok in range(V):
        assert _rejection_step(p, p, tok, u=0.999999) is None
    # disjoint: always rejected, resample lands in p's support
    p2 = torch.tensor([0.5, 0.5, 0, 0, 0, 0, 0, 0.0])
    q2 = torch.tensor([0, 0, 0.5, 0.5, 0, 0, 0, 0.0])
    for tok in (2, 3):
        r = _rejection_step(p2, q2, tok, u=0.5)
        assert r in (0, 1)
    # empirical law: proposal ~ q then rejection step => sample ~ p
    n = 20000
    counts = torch.zeros(V)
    toks = torch.multinomial(q, n, replacement=True)
    for t in toks.tolist():
        r = _rejection_step(p, q, t)
        counts[t if r is None else r] += 1
    emp = counts / n
    assert (emp - p).abs().max().item() < 0.02, (emp, p)


def test_speculative_decoding_sampled_runs():
    """temperature > 0 spec decoding: runs, respects vocab/stop/max_tokens;
    draft == target accepts aggressively (acceptance prob 1 when p == q)."""
    from mlx_cuda_distributed_pretraining_amd.inference.speculative import (
        speculative_generate_tokens)

    torch.manual_seed(1)
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs
    args = ModelArgs(hidden_size=32, intermediate_size=64, num_layers=2,
                     num_heads=2, num_kv_heads=2, vocab_size=61,
                     max_position_embeddings=128)
    model = Model(args).eval()
    prompt = [1, 5, 9]
    out = speculative_generate_tokens(model, model, prompt, max_tokens=12,
                                      k=3, temperature=0.8)
    assert 0 < len(out) <= 12
    assert all(0 <= t < model.args.vocab_size for t in out)
