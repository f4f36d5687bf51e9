"""fp8 quantization helpers (CPU; the _scaled_mm GEMM itself is GPU-only and
covered in tests/test_gpu_kernels.py)."""
import pytest
import torch

from mlx_cuda_distributed_pretraining_amd.ops.fp8 import (
    E4M3_MAX, dequantize, fp8_linear_ref, quantize_e4m3,
)


def test_quantize_roundtrip_accuracy():
    torch.manual_seed(0)
    x = torch.randn(64, 128) * 3.0
    q, s = quantize_e4m3(x)
    assert q.dtype == torch.float8_e4m3fn
    xd = dequantize(q, s, torch.float32)
    rel = (xd - x).abs().max() / x.abs().max()
    assert rel < 0.06, rel  # e4m3 has ~2-3 bits mantissa at full scale


def test_quantize_uses_full_range():
    x = torch.tensor([[1e-3, -2e-3], [3e-3, 4e-3]])
    q, s = quantize_e4m3(x)
    # amax maps to E4M3_MAX
    assert q.float().abs().max() == pytest.approx(E4M3_MAX, rel=0.05)
    assert torch.allclose(dequantize(q, s, torch.float32), x, rtol=0.08, atol=1e-6)


def test_fp8_linear_ref_close_to_exact():
    torch.manual_seed(1)
    x = torch.randn(32, 64)
    w = torch.randn(48, 64)
    y = fp8_linear_ref(x, w)
    exact = x @ w.t()
    err = (y - exact).abs().max() / exact.abs().max()
    assert err < 0.08, err


def test_model_fp8_flag_plumbed():
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    args = ModelArgs(hidden_size=32, intermediate_size=64, num_layers=1,
                     num_heads=2, num_kv_heads=2, vocab_size=50, fp8=True)
    m = Model(args)
    assert m.layers[0].attention.wqkv.fp8
    assert m.layers[0].mlp.w_down.fp8
    # CPU forward ignores fp8 (falls to F.linear) and still works
    out = m(torch.randint(0, 50, (1, 8)))
    assert out.shape == (1, 8, 50)
