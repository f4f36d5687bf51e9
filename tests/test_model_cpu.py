import torch

from mlx_cuda_distributed_pretraining_amd.models.llama import (
    Model, ModelArgs, make_prompt_cache,
)

torch.manual_seed(0)


def tiny_args(**kw):
    defaults = dict(
        hidden_size=64, intermediate_size=128, num_layers=2, num_heads=4,
        num_kv_heads=2, vocab_size=101,
    )
    defaults.update(kw)
    return ModelArgs(**defaults)


def test_forward_shape():
    m = Model(tiny_args())
    x = torch.randint(0, 101, (2, 16))
    logits = m(x)
    assert logits.shape == (2, 16, 101)


def test_untied_head_and_logit_scale():
    m = Model(tiny_args(tie_word_embeddings=False, logit_scale=2.0))
    x = torch.randint(0, 101, (1, 8))
    logits = m(x)
    assert logits.shape == (1, 8, 101)
    assert hasattr(m, "output")


def test_kv_cache_decode_matches_full_forward():
    m = Model(tiny_args())
    m.eval()
    x = torch.randint(0, 101, (1, 12))
    with torch.no_grad():
        full = m(x)
        cache = make_prompt_cache(m)
        # prefill 8, then decode 4 one at a time
        m(x[:, :8], cache=cache)
        outs = []
        for t in range(8, 12):
            outs.append(m(x[:, t : t + 1], cache=cache))
        stepped = torch.cat(outs, dim=1)
    assert torch.allclose(full[:, 8:], stepped, atol=1e-4)


def test_gradient_checkpointing_same_grads():
    args = tiny_args()
    m1 = Model(args)
    m2 = Model(args)
    m2.load_state_dict(m1.state_dict())
    for layer in m2.layers:
        layer.enable_checkpointing()
    x = torch.randint(0, 101, (2, 10))
    tgt = torch.randint(0, 101, (2, 10))
    for m in (m1, m2):
        m.train()
        loss = torch.nn.functional.cross_entropy(m(x).reshape(-1, 101), tgt.reshape(-1))
        loss.backward()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5)


def test_loss_decreases_under_training():
    m = Model(tiny_args())
    opt = torch.optim.AdamW(m.parameters(), lr=1e-2)
    x = torch.randint(0, 101, (4, 16))
    first = last = None
    for i in range(30):
        logits = m(x[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 101), x[:, 1:].reshape(-1)
        )
        opt.zero_grad()
        loss.backward()
        opt.step()
        if first is None:
            first = loss.item()
        last = loss.item()
    assert last < first * 0.7, f"loss did not decrease: {first} -> {last}"


def test_weights_save_load_roundtrip(tmp_path):
    from safetensors.torch import save_file

    m1 = Model(tiny_args())
    m2 = Model(tiny_args())
    path = tmp_path / "w.safetensors"
    save_file({k: v.contiguous() for k, v in m1.state_dict().items()}, str(path))
    m2.load_weights(str(path))
    for (k1, v1), (k2, v2) in zip(m1.state_dict().items(), m2.state_dict().items()):
        assert torch.equal(v1, v2), k1
