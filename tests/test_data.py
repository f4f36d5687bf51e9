import json

import torch

from mlx_cuda_distributed_pretraining_amd.core.config import DataConfig
from mlx_cuda_distributed_pretraining_amd.data import DataManager, TokenizerManager


def byte_tok():
    return TokenizerManager(DataConfig())


def test_byte_tokenizer_roundtrip():
    tok = byte_tok()
    text = "hello MI355X"
    ids = tok.tokenize(text)
    assert tok.detokenize(ids) == text
    assert tok.vocab_size == 256 + 3
    assert tok.PAD_TOKEN != tok.BOS_TOKEN != tok.EOS_TOKEN


def test_tokenize_doc_framing_and_truncation():
    tok = byte_tok()
    ids = tok.tokenize_doc("abc")
    assert ids[0] == tok.BOS_TOKEN and ids[-1] == tok.EOS_TOKEN
    ids = tok.tokenize_doc("x" * 100, max_length=10)
    assert len(ids) == 10
    assert ids[0] == tok.BOS_TOKEN and ids[-1] == tok.EOS_TOKEN


def test_jsonl_loading_and_batching(tmp_path):
    f = tmp_path / "train.jsonl"
    docs = [{"text": "a" * (5 + 2 * i)} for i in range(20)]  # all fit in one chunk
    f.write_text("\n".join(json.dumps(d) for d in docs))
    cfg = DataConfig(input_file=str(f), preprocessing={"max_context_size": 64, "chunk_overlap": 0})
    tok = TokenizerManager(cfg)
    dm = DataManager(cfg, tok, batch_size=4, seed=0)
    assert dm.num_batches == 5
    b = dm.generate_batch(0)
    assert b.ndim == 2 and b.shape[0] == 4
    assert b.shape[1] % 8 == 0  # padded to multiple of 8
    assert (b == tok.PAD_TOKEN).any() or b.shape[1] >= 5  # padding applied somewhere


def test_chunking_with_overlap(tmp_path):
    f = tmp_path / "train.jsonl"
    f.write_text(json.dumps({"text": "z" * 300}))
    cfg = DataConfig(input_file=str(f), preprocessing={"max_context_size": 128, "chunk_overlap": 16})
    tok = TokenizerManager(cfg)
    dm = DataManager(cfg, tok, batch_size=2)
    # 302 tokens with BOS/EOS, chunks of 128 stepping 112
    assert len(dm.docs) >= 3
    assert all(len(c) <= 128 for c in dm.docs)


def test_synthetic_batches_deterministic():
    cfg = DataConfig(synthetic=True, synthetic_vocab_size=100,
                     preprocessing={"max_context_size": 32, "chunk_overlap": 0})
    tok = TokenizerManager(cfg)
    dm = DataManager(cfg, tok, batch_size=2, seed=7)
    b1 = dm.generate_batch(3)
    b2 = dm.generate_batch(3)
    assert torch.equal(b1, b2)
    assert b1.shape == (2, 32)
    assert b1.max() < 100
    # different ranks see different data
    dm2 = DataManager(cfg, tok, batch_size=2, rank=1, world_size=2, seed=7)
    assert not torch.equal(dm2.generate_batch(3), b1)


def test_bpe_tokenizer_training(tmp_path):
    from mlx_cuda_distributed_pretraining_amd.data.tokenizer import train_bpe_tokenizer

    f = tmp_path / "corpus.jsonl"
    f.write_text("\n".join(json.dumps({"text": f"the quick brown fox {i}"}) for i in range(50)))
    out = train_bpe_tokenizer([str(f)], vocab_size=300, out_dir=str(tmp_path / "tok"))
    cfg = DataConfig(tokenizer_path=str(tmp_path / "tok"))
    tok = TokenizerManager(cfg)
    assert tok.use_external_tokenizer
    ids = tok.tokenize("the quick brown fox")
    assert len(ids) > 0
    assert "quick" in tok.detokenize(ids)


def test_synthetic_vocab_capped_to_tokenizer_vocab():
    """Synthetic ids >= the model vocab are an OOB embedding gather (a GPU
    memory fault with no traceback — the 7B debugging trail): DataManager
    caps and warns."""
    import warnings

    from mlx_cuda_distributed_pretraining_amd.data.dataset import DataManager

    class Tok:
        vocab_size = 100
        PAD_TOKEN = 0

    cfg = type("D", (), {})()
    cfg.preprocessing = {"max_context_size": 8}
    cfg.synthetic = True
    cfg.synthetic_vocab_size = 32000
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        dm = DataManager(cfg, Tok(), batch_size=2)
        assert any("capping" in str(x.message) for x in w)
    b = dm.generate_batch(0)
    assert int(b.max()) < 100
