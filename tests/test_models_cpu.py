

def test_pair_path_matches_legacy_layer_chain():
    """Model.forward's residual-fused pair chain must equal the legacy
    per-layer composition (x = layer(x)) bit-for-bit on CPU (the fused
    add+norm is the same math; the legacy path remains for static decode)."""
    import torch
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    torch.manual_seed(0)
    args = ModelArgs(hidden_size=64, intermediate_size=128, num_layers=3,
                     num_heads=4, num_kv_heads=2, vocab_size=97,
                     max_position_embeddings=64)
    m = Model(args).eval()
    tokens = torch.randint(0, 97, (2, 16))
    with torch.no_grad():
        fused = m(tokens)
        # legacy composition
        x = m.tok_embeddings(tokens)
        for layer in m.layers:
            x = layer(x, None)
        x = m.norm(x)
        legacy = x @ m.tok_embeddings.weight.t()
    assert torch.allclose(fused, legacy, atol=1e-5, rtol=1e-5), \
        (fused - legacy).abs().max()
