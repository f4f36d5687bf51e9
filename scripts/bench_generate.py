#!/usr/bin/env python3
"""Decode/serving micro-benchmark: prefill and per-token decode throughput
for a random-init model (no network; synthetic prompt tokens).

Usage: python scripts/bench_generate.py [--config configs/model-config-1b.yaml]
       [--prompt-len 512] [--decode-tokens 128] [--kv-bits 8]
"""
import argparse
import sys
import time
from pathlib import Path

import torch

REPO = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(REPO))

from mlx_cuda_distributed_pretraining_amd.core.config import Config  # noqa: E402
from mlx_cuda_distributed_pretraining_amd.inference.kv_cache import (  # noqa: E402
    make_cache, maybe_quantize_kv_cache,
)
from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--config", default=str(REPO / "configs" / "model-config-1b.yaml"))
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--decode-tokens", type=int, default=128)
    p.add_argument("--kv-bits", type=int, default=None)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--eager", action="store_true",
                   help="eager KV-cache loop instead of the hipGraph decoder")
    args = p.parse_args()

    on_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if on_gpu else "cpu")
    dtype = torch.bfloat16 if on_gpu else torch.float32

    cfg = Config.from_yaml(args.config)
    tk = cfg.data.tokenizer
    vocab = int(tk.get("normal_vocab_size", 32000)) + len(tk.get("special_tokens", {}))
    margs = ModelArgs.from_config(cfg.model, vocab)
    torch.manual_seed(0)
    model = Model(margs).to(dev, dtype).eval()

    toks = torch.randint(0, vocab, (args.batch, args.prompt_len), device=dev)

    if not args.eager and on_gpu and args.kv_bits is None and args.batch <= 8:
        # hipGraph-captured decoder (the serving path)
        from mlx_cuda_distributed_pretraining_amd.inference.static_decode import (
            GraphDecoder,
        )

        dec = GraphDecoder(model, batch=args.batch,
                           max_len=args.prompt_len + args.decode_tokens + 8)
        t0 = time.perf_counter()
        dec.prefill(toks)
        torch.cuda.synchronize()
        t_prefill = time.perf_counter() - t0
        dec.capture()
        dec.decode(8)  # warm replays
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        dec.decode(args.decode_tokens - 8)
        torch.cuda.synchronize()
        t_decode = time.perf_counter() - t0
        n_pre = args.batch * args.prompt_len
        n_dec = args.batch * (args.decode_tokens - 8)
        print(f"prefill: {n_pre} toks in {t_prefill*1e3:.1f} ms = {n_pre/t_prefill:.0f} tok/s")
        print(f"decode (hipGraph): {n_dec} toks in {t_decode*1e3:.1f} ms = "
              f"{n_dec/t_decode:.1f} tok/s ({1e3*t_decode/max(n_dec//args.batch,1):.2f} ms/step)")
        return

    def sync():
        if on_gpu:
            torch.cuda.synchronize()

    with torch.no_grad():
        # warmup
        cache = make_cache(model, "chunked")
        model(toks, cache=cache)
        sync()

        # prefill
        cache = make_cache(model, "chunked")
        t0 = time.perf_counter()
        logits = model(toks, cache=cache)
        sync()
        t_prefill = time.perf_counter() - t0

        cache = maybe_quantize_kv_cache(cache, 0, args.kv_bits)
        # decode
        cur = logits[:, -1].argmax(-1, keepdim=True)
        t0 = time.perf_counter()
        for _ in range(args.decode_tokens):
            logits = model(cur, cache=cache)
            cur = logits[:, -1].argmax(-1, keepdim=True)
        sync()
        t_decode = time.perf_counter() - t0

    n_prefill = args.batch * args.prompt_len
    n_decode = args.batch * args.decode_tokens
    print(f"prefill: {n_prefill} toks in {t_prefill*1e3:.1f} ms "
          f"= {n_prefill/t_prefill:.0f} tok/s")
    print(f"decode : {n_decode} toks in {t_decode*1e3:.1f} ms "
          f"= {n_decode/t_decode:.1f} tok/s "
          f"({1e3*t_decode/args.decode_tokens:.2f} ms/token, "
          f"kv_bits={args.kv_bits})")


if __name__ == "__main__":
    main()
