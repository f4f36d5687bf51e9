"""KV cache variants: chunked preallocation, int8/int4 group quantization,
threshold-based quantize swap, decode consistency."""
import pytest
import torch

from mlx_cuda_distributed_pretraining_amd.inference.kv_cache import (
    ChunkedKVCache, QuantizedKVCache, make_cache, maybe_quantize_kv_cache,
)
from mlx_cuda_distributed_pretraining_amd.models.llama import KVCache, Model, ModelArgs


def test_chunked_matches_simple():
    torch.manual_seed(0)
    simple, chunked = KVCache(), ChunkedKVCache(chunk=8)
    for _ in range(5):
        k = torch.randn(2, 3, 4, 16)
        v = torch.randn(2, 3, 4, 16)
        ks, vs = simple.update(k, v)
        kc, vc = chunked.update(k, v)
        assert torch.equal(ks, kc) and torch.equal(vs, vc)
        assert simple.offset == chunked.offset
    # capacity grew in chunks, not per-append
    assert chunked.k.shape[1] % 8 == 0


@pytest.mark.parametrize("bits,atol", [(8, 0.02), (4, 0.2)])
def test_quantized_cache_roundtrip(bits, atol):
    torch.manual_seed(0)
    q = QuantizedKVCache(bits=bits, group=64)
    k = torch.randn(1, 7, 2, 128)
    v = torch.randn(1, 7, 2, 128)
    kd, vd = q.update(k, v)
    assert kd.shape == k.shape
    scale = k.abs().max()
    assert (kd - k).abs().max() / scale < atol
    assert (vd - v).abs().max() / scale < atol
    # append more; previously stored rows must be stable
    k2 = torch.randn(1, 1, 2, 128)
    kd2, _ = q.update(k2, k2)
    assert torch.equal(kd2[:, :7], kd)
    assert q.offset == 8


def test_from_cache_conversion():
    torch.manual_seed(1)
    c = ChunkedKVCache()
    k = torch.randn(1, 9, 2, 64)
    c.update(k, k + 1)
    q = QuantizedKVCache.from_cache(c, bits=8, group=32)
    assert q.offset == 9
    kd, vd = q.update(torch.randn(1, 1, 2, 64), torch.randn(1, 1, 2, 64))
    assert kd.shape[1] == 10
    assert (kd[:, :9] - k).abs().max() < 0.05


def test_maybe_quantize_threshold():
    c = [ChunkedKVCache()]
    k = torch.randn(1, 4, 2, 64)
    c[0].update(k, k)
    # below threshold: unchanged
    out = maybe_quantize_kv_cache(c, quantized_kv_start=10, kv_bits=8)
    assert out[0] is c[0]
    # above threshold: swapped
    out = maybe_quantize_kv_cache(c, quantized_kv_start=2, kv_bits=8)
    assert isinstance(out[0], QuantizedKVCache)
    # idempotent
    out2 = maybe_quantize_kv_cache(out, quantized_kv_start=2, kv_bits=8)
    assert out2[0] is out[0]
    # disabled
    assert maybe_quantize_kv_cache(c, 0, None)[0] is c[0]


def _tiny_model():
    torch.manual_seed(0)
    args = ModelArgs(hidden_size=32, intermediate_size=64, num_layers=2,
                     num_heads=2, num_kv_heads=2, vocab_size=67)
    return Model(args).eval()


def test_model_decode_chunked_matches_full_forward():
    model = _tiny_model()
    toks = torch.randint(0, 67, (1, 12))
    with torch.no_grad():
        full = model(toks)
        cache = make_cache(model, kind="chunked")
        step_logits = []
        for t in range(12):
            lg = model(toks[:, t : t + 1], cache=cache)
            step_logits.append(lg[:, -1])
        inc = torch.stack(step_logits, dim=1)
    assert torch.allclose(full, inc, atol=1e-4), (full - inc).abs().max()


def test_generate_step_with_quantized_kv():
    from mlx_cuda_distributed_pretraining_amd.inference.generate import generate_step

    model = _tiny_model()
    out = list(generate_step(model, [1, 5, 9], max_tokens=8,
                             kv_bits=8, quantized_kv_start=0))
    assert len(out) == 8
    out2 = list(generate_step(model, [1, 5, 9], max_tokens=8))
    # greedy int8-KV decode should match the fp path on a tiny model
    assert out == out2


def test_chunked_trim_rewind_matches_simple():
    """inherited trim sliced capacity without moving _len — later
    updates then wrote at stale positions (speculative decoding rewind)."""
    torch.manual_seed(1)
    simple, chunked = KVCache(), ChunkedKVCache(chunk=8)
    k0, v0 = torch.randn(1, 6, 2, 16), torch.randn(1, 6, 2, 16)
    simple.update(k0, v0)
    chunked.update(k0, v0)
    simple.trim(2)
    chunked.trim(2)
    assert chunked.offset == simple.offset == 4
    k1, v1 = torch.randn(1, 3, 2, 16), torch.randn(1, 3, 2, 16)
    ks, vs = simple.update(k1, v1)
    kc, vc = chunked.update(k1, v1)
    assert torch.equal(ks, kc) and torch.equal(vs, vc)
    assert chunked.offset == simple.offset == 7


def test_quantized_trim_rewind():
    """inherited trim was a silent no-op (self.k is None)."""
    torch.manual_seed(2)
    q = QuantizedKVCache(bits=8, group=64)
    k = torch.randn(1, 6, 2, 128)
    v = torch.randn(1, 6, 2, 128)
    q.update(k, v)
    q.trim(2)
    assert q.offset == 4
    k1, v1 = torch.randn(1, 1, 2, 128), torch.randn(1, 1, 2, 128)
    kd, vd = q.update(k1, v1)
    assert kd.shape[1] == 5 and vd.shape[1] == 5
    # rewound rows really were dropped: position 4 is the NEW row
    assert torch.allclose(kd[:, 4].float(), k1[:, 0].float(), atol=0.02)
    assert torch.allclose(kd[:, :4].float(), k[:, :4].float(), atol=0.02)
