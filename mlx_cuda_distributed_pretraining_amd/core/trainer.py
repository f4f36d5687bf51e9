"""Trainer runtime.

Orchestration parity with the reference Trainer
(/root/reference/core/training.py:898-1904): seeding, device setup, model
build from config, loss with pad masking + truncation warning, validation
capped at 50 batches, checkpoint save/load/resume, the training loop with
gradient accumulation/clipping, early stopping, LR finder, per-step log
lines, sample generation during training, final metadata.

MI355X redesign:
  - one process per GPU; RCCL bucketed all-reduce overlapped with backward
    (parallel/ddp.py) instead of the reference's thread-queue mock,
  - bf16 params + fused flat AdamW (fp32 master) as the default GPU path,
  - loss/token-count scalar all-reduce to rank 0 for logging (C2).
"""
from __future__ import annotations

import csv
import math
import os
import random
import time
from pathlib import Path
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from ..data import DataManager, TokenizerManager
from ..models.llama import Model, ModelArgs
from ..ops.cross_entropy import fused_cross_entropy
from ..optim.enhanced import clip_by_global_norm, global_grad_norm
from ..optim.flat_fused import FusedFlatAdamW
from ..optim.manager import OptimizationManager
from ..parallel.ddp import DataParallelGrads
from ..parallel.dist import (
    all_reduce_scalar,
    barrier,
    broadcast_module,
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)
from ..parallel.flat import FlatParamSpace
from .checkpoint import (
    CheckpointManager,
    load_checkpoint,
    rotate_snapshots,
    save_checkpoint,
    update_metadata,
)
from .config import Config
from .logger import Logger, format_metrics


class EarlyStoppingMonitor:
    """Parity: /root/reference/core/training.py:621-668."""

    def __init__(self, patience: int = 3, min_delta: float = 0.001, mode: str = "min"):
        self.patience = patience
        self.min_delta = min_delta
        self.mode = mode
        self.best: Optional[float] = None
        self.counter = 0

    def update(self, value: float) -> bool:
        """Returns True when training should stop."""
        improved = (
            self.best is None
            or (self.mode == "min" and value < self.best - self.min_delta)
            or (self.mode == "max" and value > self.best + self.min_delta)
        )
        if improved:
            self.best = value
            self.counter = 0
        else:
            self.counter += 1
        return self.counter >= self.patience


class Trainer:
    def __init__(self, config: str | Config, for_training: bool = True, runs_root: str = "runs"):
        self.config = config if isinstance(config, Config) else Config.from_yaml(config)
        self.for_training = for_training
        self.runs_root = runs_root
        self.rank, self.world_size, self.local_rank = init_distributed(
            backend=self.config.system.distributed_backend
            if torch.cuda.is_available()
            else "gloo"
        )
        self.is_main = self.rank == 0

        self.setup_system()

        cfg = self.config
        if for_training and self.is_main:
            if not cfg.overwrite and cfg.resume is None:
                CheckpointManager.validate_unique_name(cfg.name, self.runs_root)
        barrier()
        self.run_dir, self.log_path, self.checkpoint_dir = CheckpointManager.setup_run_directory(
            cfg.name, self.runs_root
        )
        self.logger = Logger(
            self.run_dir, self.log_path, cfg.logging, is_main=self.is_main,
            use_tensorboard=cfg.logging.tensorboard,
        )
        if self.is_main and for_training:
            cfg.save_yaml(self.run_dir / "config.yaml")

        self.tokenizer = TokenizerManager(cfg.data)
        if self.is_main and for_training:
            self.tokenizer.save(self.run_dir / "tokenizer")

        self.setup_model()
        self.data_manager: Optional[DataManager] = None
        self.total_steps = 0
        self.start_step = 0
        self.total_tokens = 0
        self.validation_losses: List[tuple] = []
        self.flat_space = None
        self.optimizer = None
        self.ddp = None
        if for_training:
            self.setup_training()

    # ------------------------------------------------------------------
    def setup_system(self) -> None:
        cfg = self.config.system
        seed = cfg.seed + self.rank
        random.seed(seed)
        np.random.seed(seed)
        torch.manual_seed(seed)
        if torch.cuda.is_available() and cfg.device != "cpu":
            torch.cuda.set_device(self.local_rank)
            torch.cuda.manual_seed_all(seed)
            self.device = torch.device("cuda", self.local_rank)
        else:
            self.device = torch.device("cpu")
        if cfg.precision == "bfloat16":
            self.param_dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        elif cfg.precision == "float16":
            self.param_dtype = torch.float16 if self.device.type == "cuda" else torch.float32
        else:
            self.param_dtype = torch.float32

    def setup_model(self) -> None:
        args = ModelArgs.from_config(self.config.model, vocab_size=self.tokenizer.vocab_size)
        if self.config.model.architecture not in ("llama", "llama_standard"):
            raise ValueError(
                f"Unknown architecture {self.config.model.architecture!r} (supported: llama, llama_standard)"
            )
        if self.config.model.architecture == "llama_standard":
            args.attention_type = "simple"
        self.model_args = args
        # Multi-billion-parameter models build directly ON the training
        # device: CPU weight init is single-threaded torch.normal_ and takes
        # minutes at 7B scale, while the same init on the GPU is sub-second
        # (288 GB HBM holds the whole model anyway). Smaller models keep the
        # plain CPU-init path.
        est_params = (
            args.vocab_size * args.hidden_size
            + args.num_layers
            * (args.hidden_size * (args.num_heads + 2 * args.num_kv_heads) * args.head_dim
               + args.hidden_size * args.num_heads * args.head_dim
               + 3 * args.hidden_size * args.intermediate_size)
        )
        if est_params > 3_000_000_000 and self.device.type == "cuda":
            try:
                with torch.device(self.device):
                    self.model = Model(args)
            except Exception:
                self.model = Model(args)  # CPU init fallback (slow but safe)
        else:
            self.model = Model(args)
        self.model = self.model.to(device=self.device, dtype=self.param_dtype)
        if self.config.system.gradient_checkpointing:
            ratio = self.config.system.gradient_checkpointing_ratio
            n = int(len(self.model.layers) * ratio)
            for layer in self.model.layers[:n]:
                layer.enable_checkpointing()
        if self.config.data.weight_path:
            self.model.load_weights(self.config.data.weight_path)
        broadcast_module(self.model)
        # Tensor parallelism (parallel/tp.py): shard the broadcast model.
        # The reference's model_parallel flags were a logged placeholder
        # (/root/reference/core/training.py:1178-1193); here they are real.
        self.tp_world, self.dp_world = 1, self.world_size
        self.tp_rank, self.dp_rank = 0, self.rank
        self.tp_pg = self.dp_pg = None
        self.sp = False
        if self.config.system.model_parallel and self.world_size > 1:
            mp_size = int(self.config.system.model_parallel_size or self.world_size)
            from ..parallel.tp import apply_tensor_parallel, init_tp_mesh

            # DPxTP mesh: TP groups are adjacent ranks (one xGMI node per
            # replica), DP groups stride across replicas
            self.tp_rank, self.dp_rank, self.tp_pg, self.dp_pg = init_tp_mesh(
                self.rank, self.world_size, mp_size
            )
            self.sp = bool(self.config.system.sequence_parallel)
            apply_tensor_parallel(self.model, self.tp_rank, mp_size,
                                  sequence_parallel=self.sp)
            self.tp_world = mp_size
            self.dp_world = self.world_size // mp_size
            if self.is_main:
                self.logger.log(
                    f"Mesh: TP={self.tp_world} x DP={self.dp_world}"
                    + (" + sequence parallel" if self.sp else "")
                    + " (head/intermediate sharded, 1 all-reduce per sublayer)"
                )
        self.logger.log_model_summary(self.model)

    def setup_training(self) -> None:
        cfg = self.config
        hp = cfg.training.hyperparameters
        self.batch_size = int(hp.get("batch_size", 16))
        self.grad_accum_steps = int(hp.get("gradient_accumulation_steps", 1))
        self.max_grad_norm = float(hp.get("gradient_clip", hp.get("max_grad_norm", 0.0)) or 0.0)

        # TP replicas share data; DP replicas shard it (mesh coordinates)
        self.data_manager = DataManager(
            cfg.data, self.tokenizer, self.batch_size,
            rank=self.dp_rank, world_size=self.dp_world, seed=cfg.system.seed,
        )
        if cfg.training.epochs is not None and self.data_manager.num_batches:
            # Data is sharded over the DP group only (TP replicas share
            # batches), so epochs divide by dp_world, not world_size.
            self.steps_per_epoch = max(
                self.data_manager.num_batches // self.dp_world, 1
            )
            self.total_steps = self.steps_per_epoch * cfg.training.epochs
        else:
            self.steps_per_epoch = None
            self.total_steps = int(hp.get("iters", 1000))

        self.opt_manager = OptimizationManager(cfg.training, self.total_steps)
        self.lr_schedule = self.opt_manager.create_scheduler()

        opt_name = str((cfg.training.optimization or {}).get("optimizer", "adamw")).lower()
        self.use_fused = (
            self.device.type == "cuda"
            and opt_name in ("adamw", "adam")
            and not cfg.system.model_parallel
        )
        self.flat_space: Optional[FlatParamSpace] = None
        self.ddp: Optional[DataParallelGrads] = None
        if self.use_fused:
            # copy-mode grads: no autograd read-modify-write add per param,
            # no per-step flat zero_ (FusedFlatAdamW reads flat_grad only)
            self.flat_space = FlatParamSpace(self.model, grad_mode="copy")
            zero1 = cfg.system.zero_optimization_level >= 1
            if not zero1:
                self.ddp = DataParallelGrads(self.flat_space, bucket_mb=cfg.system.bucket_mb)
            self.optimizer = FusedFlatAdamW(
                self.flat_space,
                lr=self.opt_manager.learning_rate,
                betas=tuple((cfg.training.optimization or {}).get("betas", (0.9, 0.999))),
                eps=float((cfg.training.optimization or {}).get("eps", 1e-8)),
                weight_decay=self.opt_manager.weight_decay,
                max_grad_norm=self.max_grad_norm,
                zero1=zero1,
            )
        else:
            self.flat_space = FlatParamSpace(self.model)
            # grads all-reduce over the DP group only: TP peers hold
            # DIFFERENT shards and their grads are already exact
            if self.tp_world > 1:
                self.ddp = (DataParallelGrads(self.flat_space,
                                              bucket_mb=cfg.system.bucket_mb,
                                              process_group=self.dp_pg)
                            if self.dp_world > 1 else None)
            else:
                self.ddp = DataParallelGrads(self.flat_space, bucket_mb=cfg.system.bucket_mb)
            self.optimizer = self.opt_manager.create_optimizer(self.model)


        es = cfg.training.early_stopping or {}
        self.early_stopping = (
            EarlyStoppingMonitor(
                patience=int(es.get("patience", 3)),
                min_delta=float(es.get("min_delta", 0.001)),
                mode=str(es.get("mode", "min")),
            )
            if es.get("enabled")
            else None
        )

    # ------------------------------------------------------------------
    def _tp_clip(self) -> None:
        """Global-norm clip under TP: norm^2 = sum of sharded-param grads over
        ALL ranks + replicated params counted ONCE; every rank applies the
        same scale so replicas stay bit-identical."""
        import torch.distributed as dist

        sh = torch.zeros((), dtype=torch.float32, device=self.device)
        rep = torch.zeros((), dtype=torch.float32, device=self.device)
        for p in self.model.parameters():
            if p.grad is None:
                continue
            ss = p.grad.float().pow(2).sum()
            if getattr(p, "_tp_sharded", False):
                sh += ss
            else:
                rep += ss
        if is_distributed():
            dist.all_reduce(sh, group=self.tp_pg)
        norm = (sh + rep).sqrt()
        scale = self.max_grad_norm / (norm + 1e-6)
        if float(scale) < 1.0:
            for p in self.model.parameters():
                if p.grad is not None:
                    p.grad.mul_(scale)

    def compute_loss(self, inputs: torch.Tensor, targets: torch.Tensor):
        mpe = self.model_args.max_position_embeddings
        if mpe and inputs.shape[1] > mpe:
            if self.is_main:
                self.logger.log(
                    f"Warning: sequence length {inputs.shape[1]} > max_position_embeddings {mpe}; truncating",
                    console=False,
                )
            inputs = inputs[:, :mpe]
            targets = targets[:, :mpe]
        logits = self.model(inputs)
        vp0 = getattr(self.model, "_vp_vocab0", -1)
        if vp0 >= 0 and self.model.training:
            # vocab-parallel CE over the sharded lm head (parallel/tp.py):
            # the full [N, V] logits replica never materializes
            from ..parallel.tp import vocab_parallel_cross_entropy

            loss, ntok = vocab_parallel_cross_entropy(
                logits.reshape(-1, logits.shape[-1]),
                targets.reshape(-1), vp0,
                ignore_index=self.tokenizer.PAD_TOKEN,
            )
            aux = getattr(self.model, "aux_loss", None)
            if aux is not None:
                loss = loss + self.model_args.router_aux_loss_coef * aux
            return loss, ntok
        if self.sp and self.model.training and logits.shape[1] != targets.shape[1]:
            # sequence parallelism: logits are S-sharded; shard-local CE,
            # then a differentiable sum over the TP group (identity backward
            # keeps each rank's local grads unscaled)
            from ..parallel.tp import reduce_from_tp

            s_loc = logits.shape[1]
            tgt = targets[:, self.tp_rank * s_loc:(self.tp_rank + 1) * s_loc]
            loss_l, ntok_l = fused_cross_entropy(
                logits.reshape(-1, logits.shape[-1]),
                tgt.reshape(-1),
                ignore_index=self.tokenizer.PAD_TOKEN,
            )
            loss_sum = reduce_from_tp((loss_l * ntok_l.float()).unsqueeze(0)).squeeze(0)
            ntok_g = ntok_l.float().clone()
            if is_distributed():
                import torch.distributed as dist

                dist.all_reduce(ntok_g, group=self.tp_pg)  # THIS replica's tokens
            loss = loss_sum / ntok_g
            # MoE under SP: aux is computed on the gathered tokens
            # (identical per rank, pre-divided by ep_world — parallel/tp.py)
            aux = getattr(self.model, "aux_loss", None)
            if aux is not None and self.model.training:
                loss = loss + self.model_args.router_aux_loss_coef * aux
            return loss, ntok_g.long()
        loss, ntok = fused_cross_entropy(
            logits.reshape(-1, logits.shape[-1]),
            targets.reshape(-1),
            ignore_index=self.tokenizer.PAD_TOKEN,
        )
        # MoE: add the router load-balance loss (models/llama.py MoE)
        aux = getattr(self.model, "aux_loss", None)
        if aux is not None and self.model.training:
            loss = loss + self.model_args.router_aux_loss_coef * aux
        return loss, ntok

    def train_step(self, step: int) -> tuple:
        """One optimizer step (with gradient accumulation). Returns
        (loss_detached, ntokens). When ``logging.log_step_breakdown`` is on,
        fills self.last_breakdown = (data_ms, compute_ms, optim_ms) — CUDA
        events, resolved lazily (synchronizes only when read at log time)."""
        lr = self.lr_schedule(step)
        breakdown = bool(self.config.logging.log_step_breakdown)
        on_gpu = self.device.type == "cuda"
        if breakdown and on_gpu:
            ev = [torch.cuda.Event(enable_timing=True) for _ in range(3)]
        t_host0 = time.perf_counter()
        t_data = 0.0
        if self.ddp is not None:
            self.ddp.require_reduce = False
        total_loss = None
        total_tok = None
        if self.flat_space is not None:
            self.flat_space.zero_grad()
        else:
            self.model.zero_grad(set_to_none=False)
        for micro in range(self.grad_accum_steps):
            if self.ddp is not None and micro == self.grad_accum_steps - 1:
                self.ddp.require_reduce = True
            t_d0 = time.perf_counter()
            batch = self.data_manager.generate_batch(
                step * self.grad_accum_steps + micro
            ).to(self.device, non_blocking=True)
            t_data += time.perf_counter() - t_d0
            if breakdown and on_gpu and micro == 0:
                ev[0].record()
            inputs, targets = batch[:, :-1], batch[:, 1:]
            loss, ntok = self.compute_loss(inputs, targets)
            (loss / self.grad_accum_steps).backward()
            ld = loss.detach()
            total_loss = ld if total_loss is None else total_loss + ld
            total_tok = ntok if total_tok is None else total_tok + ntok
        if self.ddp is not None:
            self.ddp.finalize()
        if self.sp:
            from ..parallel.tp import sp_allreduce_replicated_grads

            sp_allreduce_replicated_grads(self.model)
        elif self.tp_world > 1:
            # EP MoE routers have partial grads (each rank backprops only its
            # local experts' gate terms) — sum them over the group
            from ..parallel.tp import ep_allreduce_router_grads

            ep_allreduce_router_grads(self.model)
        if breakdown and on_gpu:
            ev[1].record()

        if isinstance(self.optimizer, FusedFlatAdamW):
            self.optimizer.step(lr=lr)
        else:
            if self.max_grad_norm > 0 and self.tp_world > 1:
                self._tp_clip()
            elif self.max_grad_norm > 0 and not getattr(self.optimizer, "max_grad_norm", 0):
                clip_by_global_norm(
                    [p for p in self.model.parameters() if p.requires_grad],
                    self.max_grad_norm,
                )
            for group in self.optimizer.param_groups:
                group["lr"] = lr
            sub = [getattr(self.optimizer, "matrix_opt", None), getattr(self.optimizer, "non_matrix_opt", None)]
            for o in sub:
                if o is not None:
                    for group in o.param_groups:
                        group["lr"] = lr
            self.optimizer.step()
        if breakdown:
            if on_gpu:
                ev[2].record()
                self.last_breakdown = (t_data, ev, t_host0)
            else:
                self.last_breakdown = (t_data, None, t_host0)
        return total_loss / self.grad_accum_steps, total_tok

    def format_breakdown(self) -> str:
        """Resolve the last step's phase timings (synchronizes the GPU)."""
        if not getattr(self, "last_breakdown", None):
            return ""
        t_data, ev, t_host0 = self.last_breakdown
        if ev is not None:
            ev[2].synchronize()
            compute_ms = ev[0].elapsed_time(ev[1])
            optim_ms = ev[1].elapsed_time(ev[2])
        else:
            total = (time.perf_counter() - t_host0) * 1000
            compute_ms, optim_ms = total - t_data * 1000, 0.0
        return f" | data_ms={t_data*1e3:.1f} | compute_ms={compute_ms:.1f} | optim_ms={optim_ms:.1f}"

    # ------------------------------------------------------------------
    @torch.no_grad()
    def validate(self, max_batches: int = 50) -> Optional[float]:
        if self.data_manager is None:
            return None
        n = min(self.data_manager.num_validation_batches, max_batches)
        if n == 0:
            return None
        self.model.eval()
        total = torch.zeros((), dtype=torch.float32, device=self.device)
        total_tok = torch.zeros((), dtype=torch.float32, device=self.device)
        for i in range(n):
            batch = self.data_manager.generate_validation_batch(i)
            if batch is None:
                break
            batch = batch.to(self.device)
            loss, ntok = self.compute_loss(batch[:, :-1], batch[:, 1:])
            total += loss.float() * ntok.float()
            total_tok += ntok.float()
        self.model.train()
        total = all_reduce_scalar(total)
        total_tok = all_reduce_scalar(total_tok)
        if total_tok.item() == 0:
            return None
        return (total / total_tok).item()

    # ------------------------------------------------------------------
    def save_checkpoint(self, tag: str, val_loss: Optional[float] = None) -> None:
        if self.tp_world > 1:
            self._save_checkpoint_tp(tag, val_loss)
            return
        if not self.is_main:
            barrier()
            return
        base = str(self.checkpoint_dir / f"step_{tag}")
        if isinstance(self.optimizer, FusedFlatAdamW):
            opt_state = self.optimizer.state_dict()
        else:
            opt_state = self.optimizer.state_dict()
        training_state = {
            "step": self.current_step,
            "total_tokens": int(self.total_tokens),
            "validation_losses": self.validation_losses,
            "val_ptr": getattr(self.data_manager, "val_ptr", 0),
        }
        save_checkpoint(base, self.model, opt_state, training_state)
        update_metadata(
            self.run_dir,
            {"tag": str(tag), "step": self.current_step, "val_loss": val_loss,
             "path": base, "time": time.time()},
        )
        if self.config.logging.max_snapshots > 0:
            rotate_snapshots(self.checkpoint_dir, self.config.logging.max_snapshots)
        barrier()

    def _save_checkpoint_tp(self, tag: str, val_loss: Optional[float]) -> None:
        """TP runs: one dp_rank==0 rank per TP shard saves its own
        model/optimizer triple at ``step_<tag>_tp<r>``; the global main also
        writes a plain ``step_<tag>_state.json`` marker (holds tp_world) so
        ``latest_checkpoint`` / --auto-resume keep working. Snapshot rotation
        skips the shard files (markers only) — full rotation is a ROADMAP
        item."""
        import json as _json

        base = str(self.checkpoint_dir / f"step_{tag}")
        training_state = {
            "step": self.current_step,
            "total_tokens": int(self.total_tokens),
            "validation_losses": self.validation_losses,
            "val_ptr": getattr(self.data_manager, "val_ptr", 0),
            "tp_world": self.tp_world,
            "vocab_parallel": getattr(self.model, "_vp_vocab0", -1) >= 0,
        }
        if self.dp_rank == 0:
            save_checkpoint(f"{base}_tp{self.tp_rank}", self.model,
                            self.optimizer.state_dict(), training_state)
        if self.is_main:
            Path(f"{base}_state.json").write_text(_json.dumps(training_state))
            update_metadata(
                self.run_dir,
                {"tag": str(tag), "step": self.current_step, "val_loss": val_loss,
                 "path": base, "tp_world": self.tp_world, "time": time.time()},
            )
        barrier()

    def load_checkpoint(self, checkpoint_base: str, reset_optimizer: bool = False,
                        reset_training_state: bool = False) -> None:
        if self.tp_world > 1:
            sharded = f"{checkpoint_base}_tp{self.tp_rank}"
            if not Path(f"{sharded}_state.json").exists():
                raise FileNotFoundError(
                    f"TP run (degree {self.tp_world}) needs per-shard checkpoint "
                    f"files ({sharded}_*); re-sharding a different layout is not "
                    f"implemented (ROADMAP)"
                )
            checkpoint_base = sharded
        opt_state, training_state = load_checkpoint(checkpoint_base, self.model,
                                                    map_location=str(self.device))
        if self.tp_world > 1 and training_state.get("tp_world") not in (None, self.tp_world):
            raise ValueError(
                f"checkpoint was saved at tp_world={training_state.get('tp_world')}, "
                f"this run is tp_world={self.tp_world}"
            )
        if self.flat_space is not None:
            # re-sync the flat buffer with the freshly loaded param data
            for n, off, numel, shape in self.flat_space.segments:
                p = self.flat_space.name_to_param[n]
                # p.data already IS the flat view; load_state_dict copied into it
            if isinstance(self.optimizer, FusedFlatAdamW):
                self.optimizer.master.copy_(
                    self.flat_space.flat_param[
                        self.optimizer.shard_start : self.optimizer.shard_end
                    ].float()
                )
        if opt_state is not None and not reset_optimizer and self.for_training:
            try:
                self.optimizer.load_state_dict(opt_state)
            except Exception as e:
                self.logger.log(f"Warning: optimizer state not restored ({e}); starting fresh")
        if not reset_training_state:
            self.start_step = int(training_state.get("step", 0))
            self.total_tokens = int(training_state.get("total_tokens", 0))
            self.validation_losses = list(training_state.get("validation_losses", []))
        # DP replicas re-sync from rank 0; TP shards hold DIFFERENT weights
        # by design and each already loaded its own file — a world broadcast
        # would overwrite every shard with tp0's.
        if self.tp_world == 1:
            broadcast_module(self.model)

    # ------------------------------------------------------------------
    def run_learning_rate_finder(self) -> Dict[str, Any]:
        """Exponential LR sweep (parity: /root/reference/core/training.py:671-761,
        :1480-1537). Writes lr_finder.csv in the run dir and returns the
        suggestion."""
        lf = self.config.training.lr_finder or {}
        min_lr = float(lf.get("min_lr", 1e-7))
        max_lr = float(lf.get("max_lr", 1.0))
        num_steps = int(lf.get("num_steps", 100))
        mult = (max_lr / min_lr) ** (1.0 / max(num_steps - 1, 1))
        records = []
        for i in range(num_steps):
            lr = min_lr * (mult**i)
            self.lr_schedule = lambda step, _lr=lr: _lr
            loss, _ = self.train_step(i)
            loss_v = float(all_reduce_scalar(loss.float(), "mean").item())
            records.append((lr, loss_v))
            if not math.isfinite(loss_v) or (records and loss_v > 4 * min(r[1] for r in records)):
                break
        # steepest descent suggestion
        best_lr = records[0][0]
        best_slope = 0.0
        for j in range(1, len(records)):
            dlr = math.log(records[j][0]) - math.log(records[j - 1][0])
            slope = (records[j][1] - records[j - 1][1]) / dlr
            if slope < best_slope:
                best_slope = slope
                best_lr = records[j][0]
        if self.is_main:
            with open(self.run_dir / "lr_finder.csv", "w", newline="") as f:
                w = csv.writer(f)
                w.writerow(["lr", "loss"])
                w.writerows(records)
            try:  # loss-vs-lr plot (reference core/training.py:717-746)
                import matplotlib

                matplotlib.use("Agg")
                import matplotlib.pyplot as plt

                fig, ax = plt.subplots(figsize=(8, 5))
                ax.plot([r[0] for r in records], [r[1] for r in records], "o-")
                ax.axvline(best_lr, color="r", ls="--", label=f"suggested {best_lr:.2e}")
                ax.set_xscale("log"); ax.set_xlabel("learning rate"); ax.set_ylabel("loss")
                ax.legend(); ax.grid(alpha=0.3)
                fig.savefig(self.run_dir / "lr_finder.png", dpi=120)
                plt.close(fig)
            except Exception:
                pass
            self.logger.log(f"LR finder suggestion: {best_lr:.3e}")
        return {"suggested_lr": best_lr, "records": records}

    # ------------------------------------------------------------------
    def generate_sample(self, prompt: str = "", max_tokens: int = 64) -> str:
        from ..inference.generate import generate

        try:
            text, _stats = generate(
                self.model, self.tokenizer, prompt, max_tokens=max_tokens, temperature=0.8
            )
            return text
        except Exception as e:  # sampling must never kill training
            return f"<generation failed: {e}>"

    # ------------------------------------------------------------------
    def train(self) -> None:
        cfg = self.config
        steps_cfg = cfg.logging.steps
        log_interval = int(steps_cfg.get("logging_interval", 1))
        ckpt_interval = int(steps_cfg.get("checkpoint_interval", 0))
        val_interval = int(steps_cfg.get("validation_interval", 0))

        # optional WebSocket stats mesh (reference stats_server/stats_client)
        stats_client = None
        stats_collector = None
        if cfg.logging.stats_url:
            try:
                from ..utils.stats_client import StatsClient, WorkerMetricsCollector

                stats_client = StatsClient(
                    cfg.logging.stats_url, worker_id=f"rank{self.rank}",
                    info={"run": cfg.name, "world_size": self.world_size},
                )
                stats_client.start()
                stats_collector = WorkerMetricsCollector(
                    f"rank{self.rank}", device=self.local_rank
                )
            except Exception as e:  # stats are best-effort
                self.logger.log(f"stats mesh unavailable: {e}")

        self.current_step = 0
        if cfg.resume is not None:
            self.load_checkpoint(
                cfg.resume.checkpoint,
                reset_optimizer=cfg.resume.reset_optimizer,
                reset_training_state=cfg.resume.reset_training_state,
            )
            self.logger.log(f"Resumed from {cfg.resume.checkpoint} at step {self.start_step}")
            if self.start_step > 0:
                skipped = self.data_manager.fast_forward(
                    self.start_step * self.grad_accum_steps
                )
                if skipped:
                    self.logger.log(
                        f"Streaming source fast-forwarded past {skipped} batches"
                    )

        if (cfg.training.lr_finder or {}).get("enabled"):
            self.run_learning_rate_finder()
            return

        self.model.train()
        start_time = time.time()
        step_t0 = time.time()
        val_loss = None
        stop = False
        try:
            self._train_loop(cfg, log_interval, ckpt_interval, val_interval,
                             start_time, step_t0, val_loss, stop,
                             stats_client, stats_collector)
        except Exception as e:
            # Failure recovery: on any hot-loop failure
            # (RCCL timeout/abort included — init_distributed sets a
            # collective timeout), try to save an emergency checkpoint so
            # the run can resume from the last completed step.
            self.logger.log(f"FATAL at step {self.current_step}: {e!r}; "
                            f"saving emergency checkpoint")
            try:
                self.save_checkpoint(f"emergency_{self.current_step}")
            except Exception as e2:  # a dead process group must not mask e
                self.logger.log(f"emergency checkpoint failed: {e2!r}")
            if stats_client is not None:
                stats_client.stop()
            raise

    def _train_loop(self, cfg, log_interval, ckpt_interval, val_interval,
                    start_time, step_t0, val_loss, stop,
                    stats_client, stats_collector) -> None:
        for step in range(self.start_step, self.total_steps):
            loss, ntok = self.train_step(step)
            self.current_step = step + 1  # completed steps

            loss_g = all_reduce_scalar(loss.float(), "mean")
            ntok_g = all_reduce_scalar(ntok.float())
            loss_v = float(loss_g.item())
            # TP replicas process the SAME tokens: count each token once
            ntok_v = int(ntok_g.item()) // self.tp_world
            self.total_tokens += ntok_v

            if stats_collector is not None:
                now = time.time()
                stats_client.send_stats(
                    stats_collector.update(step + 1, loss_v, ntok_v, now - step_t0)
                )
                step_t0 = now

            if val_interval and (step + 1) % val_interval == 0:
                val_loss = self.validate()
                if val_loss is not None:
                    self.validation_losses.append((step + 1, val_loss))
                    if self.early_stopping is not None and self.early_stopping.update(val_loss):
                        self.logger.log(f"Early stopping at step {step + 1} (val_loss={val_loss:.4f})")
                        stop = True
                if cfg.logging.log_samples and self.is_main:
                    for i in range(cfg.logging.log_samples_count):
                        self.logger.log(f"Sample {i}: {self.generate_sample()!r}", console=False)

            if log_interval and (step + 1) % log_interval == 0:
                line = format_metrics(
                    step + 1, loss_v, ntok_v, self.total_tokens, start_time,
                    self.lr_schedule(step),
                    val_loss=val_loss,
                    metrics_flags=cfg.logging.metrics,
                    epochs=cfg.training.epochs,
                    steps_per_epoch=self.steps_per_epoch,
                    grad_accum_steps=self.grad_accum_steps,
                    # TP replicas share data: only DP multiplies the batch
                    effective_batch_size=self.batch_size
                    * self.grad_accum_steps
                    * self.dp_world,
                )
                if cfg.logging.log_step_breakdown:
                    line += self.format_breakdown()
                if cfg.logging.log_gradient_norm and self.flat_space is not None:
                    gnorm = float(self.flat_space.flat_grad.float().norm().item())
                    line += f" | grad_norm={gnorm:.3e}"
                self.logger.log_metrics_line(step + 1, line)
                self.logger.tb_scalars(step + 1, {"train/loss": loss_v, "train/lr": self.lr_schedule(step)})
                if cfg.logging.log_memory_usage:
                    self.logger.log_memory_usage()
                val_loss = None

            if ckpt_interval and (step + 1) % ckpt_interval == 0:
                self.save_checkpoint(str(step + 1), val_loss)
            if stop:
                break

        final_val = self.validate()
        if final_val is not None:
            self.validation_losses.append((self.current_step, final_val))
        self.save_checkpoint("final", final_val)
        self.logger.log(
            f"Training complete: {self.total_tokens} tokens, final val_loss="
            f"{final_val if final_val is not None else float('nan')}"
        )
        if stats_client is not None:
            stats_client.stop()
        self.logger.close()
