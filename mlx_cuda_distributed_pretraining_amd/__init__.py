"""MI355X-native LLM pretraining framework.

A from-scratch rebuild of the capabilities of
``arthurcolle/mlx-cuda-distributed-pretraining`` (an MLX/Metal framework)
for AMD Instinct MI355X (gfx950, CDNA4):

- PyTorch-ROCm orchestration, hand-written HIP kernels for every hot op
  (FlashAttention fwd/bwd, RMSNorm, RoPE, SwiGLU, fused cross-entropy,
  fused multi-tensor AdamW/Lion/SGD, Muon Newton-Schulz, sampling),
- RCCL (torch.distributed "nccl" backend on ROCm) data-parallel gradient
  all-reduce over xGMI, bucketed and overlapped with backward,
- the reference's YAML config schema, runs/ checkpoint layout, log-line
  format, generation/ export tooling kept compatible.

Layer map (mirrors SURVEY.md section 1):
  ops/        L1  HIP kernels + torch reference fallbacks (CPU)
  models/     L2  Llama model family
  optim/      L3  optimizer zoo (AdamW/SGD/Lion enhanced, Muon, Shampoo, Hybrid)
  core/       L4  Trainer runtime, config schema, checkpointing, logging
  parallel/   L5  RCCL data-parallel engine (bucketed all-reduce, ZeRO-1)
  data/       L5  tokenizer + data pipeline
  inference/  L6  KV-cached generation, samplers, beam search
  utils/      L8  plotting / monitoring / stats
"""

__version__ = "0.1.0"
