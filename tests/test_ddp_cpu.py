"""Multi-process data-parallel correctness on CPU (gloo, world_size=2):
the bucketed all-reduce path must produce the same gradients as a
single-process run on the concatenated batch."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _tiny_args():
    return ModelArgs(hidden_size=32, intermediate_size=64, num_layers=2,
                     num_heads=2, num_kv_heads=2, vocab_size=67)


def _make_batches(seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, 67, (4, 16), generator=g)


def _single_process_grads():
    torch.manual_seed(0)
    model = Model(_tiny_args())
    batch = _make_batches()
    logits = model(batch[:, :-1])
    loss = torch.nn.functional.cross_entropy(
        logits.reshape(-1, 67), batch[:, 1:].reshape(-1)
    )
    loss.backward()
    return {n: p.grad.clone() for n, p in model.named_parameters()}, \
           {n: p.detach().clone() for n, p in model.named_parameters()}


def _worker(rank, world_size, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from mlx_cuda_distributed_pretraining_amd.parallel.ddp import DataParallelGrads
        from mlx_cuda_distributed_pretraining_amd.parallel.dist import broadcast_module
        from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

        torch.manual_seed(0)  # same init on both ranks
        model = Model(_tiny_args())
        broadcast_module(model)
        space = FlatParamSpace(model)
        ddp = DataParallelGrads(space, bucket_mb=1)

        full = _make_batches()
        batch = full[rank * 2 : (rank + 1) * 2]  # each rank half the batch
        logits = model(batch[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 67), batch[:, 1:].reshape(-1)
        )
        loss.backward()
        ddp.finalize()
        if rank == 0:
            # send numpy copies: CPU-tensor fd-passing over the spawn queue can
            # race with worker exit (ConnectionResetError in recvfds)
            grads = {n: p.grad.numpy().copy() for n, p in model.named_parameters()}
            result_q.put(grads)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_ddp_grads_match_single_process():
    expected, _params = _single_process_grads()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = q.get()
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    # Each rank computed mean-loss over its half; all-reduce-mean of grads
    # == grad of mean over both halves == single-process grad over full batch
    # (equal token counts per rank).
    for n, g in expected.items():
        assert torch.allclose(torch.from_numpy(got[n]), g, atol=1e-5), \
            f"grad mismatch on {n}"


def _worker_zero1(rank, world_size, port, result_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        from mlx_cuda_distributed_pretraining_amd.optim.flat_fused import FusedFlatAdamW
        from mlx_cuda_distributed_pretraining_amd.parallel.dist import broadcast_module
        from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

        torch.manual_seed(0)
        model = Model(_tiny_args())
        broadcast_module(model)
        space = FlatParamSpace(model)
        opt = FusedFlatAdamW(space, lr=1e-2, weight_decay=0.0, zero1=True)

        full = _make_batches()
        batch = full[rank * 2 : (rank + 1) * 2]
        logits = model(batch[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 67), batch[:, 1:].reshape(-1)
        )
        loss.backward()
        opt.step()
        if rank == 0:
            result_q.put(space.flat_param.numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_zero1_step_runs_and_syncs():
    """ZeRO-1: reduce-scatter + shard update + all-gather must run and leave
    all ranks with finite, synchronized params."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_worker_zero1, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    flat = torch.from_numpy(q.get())
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    assert torch.isfinite(flat).all()


def test_flat_copy_mode_matches_views_mode():
    """grad_mode='copy' must produce the same flat_grad as 'views' (incl.
    gradient-accumulation add on micro-step > 0)."""
    from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

    def run(mode):
        torch.manual_seed(0)
        model = Model(_tiny_args())
        space = FlatParamSpace(model, grad_mode=mode)
        batch = _make_batches()
        for micro in range(2):  # two micro-steps: copy then accumulate
            logits = model(batch[micro * 2 : micro * 2 + 2, :-1])
            loss = torch.nn.functional.cross_entropy(
                logits.reshape(-1, 67), batch[micro * 2 : micro * 2 + 2, 1:].reshape(-1)
            )
            loss.backward()
        return space.flat_grad.clone()

    g_views = run("views")
    g_copy = run("copy")
    assert torch.allclose(g_views, g_copy, atol=1e-6), (g_views - g_copy).abs().max()


def test_flat_copy_mode_second_step_fresh():
    """zero_grad() in copy mode resets the seen-set: the next step's grads
    REPLACE (not accumulate onto) the previous step's."""
    from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

    torch.manual_seed(0)
    model = Model(_tiny_args())
    space = FlatParamSpace(model, grad_mode="copy")
    batch = _make_batches()

    def one_step():
        space.zero_grad()
        logits = model(batch[:2, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 67), batch[:2, 1:].reshape(-1)
        )
        loss.backward()
        return space.flat_grad.clone()

    g1 = one_step()
    g2 = one_step()
    assert torch.allclose(g1, g2, atol=1e-6)
