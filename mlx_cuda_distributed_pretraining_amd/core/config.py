"""YAML config schema.

Feature-parity with the reference dataclass schema
(/root/reference/core/training.py:52-167): the same top-level sections
(name / overwrite / data / model / training / logging / system / resume)
and the same nested keys, so the reference's YAML configs parse unchanged.

MI355X extensions (all optional, default off):
  system.distributed_backend: "nccl" (RCCL) | "gloo"
  system.bucket_mb:           gradient all-reduce bucket size (default 50)
  system.zero_optimization_level: 1 enables optimizer-state sharding
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml


@dataclass
class DataConfig:
    input_file: Optional[str] = None
    preprocessing: Dict[str, int] = field(
        default_factory=lambda: {"max_context_size": 1024, "chunk_overlap": 0}
    )
    tokenizer: Dict[str, Any] = field(
        default_factory=lambda: {
            "normal_vocab_size": 256,
            "special_tokens": {"pad": "<pad>", "bos": "<bos>", "eos": "<eos>"},
        }
    )
    tokenizer_path: Optional[str] = None
    validation_file: Optional[str] = None
    weight_path: Optional[str] = None
    # MI355X extension: synthetic data of a given shape (no-network benches).
    synthetic: bool = False
    synthetic_vocab_size: int = 32000
    # Streaming shard pipeline (reference fineweb_stream*.py): e.g.
    # {"source": "/data/shards", "max_cache_gb": 10, "cache_dir": "/tmp/cache"}
    streaming: Optional[Dict[str, Any]] = None


@dataclass
class ModelConfig:
    architecture: str = "llama"
    dimensions: Dict[str, int] = field(
        default_factory=lambda: {"hidden_size": 128, "intermediate_size": 256, "num_layers": 4}
    )
    attention: Dict[str, Any] = field(
        default_factory=lambda: {
            "num_heads": 8,
            "num_kv_heads": None,
            "head_dim": None,
            "max_position_embeddings": None,
        }
    )
    normalization: Dict[str, float] = field(default_factory=lambda: {"rms_norm_eps": 1e-5})
    rope: Dict[str, Any] = field(
        default_factory=lambda: {"theta": 10000, "traditional": False, "scaling": None}
    )
    misc: Dict[str, Any] = field(
        default_factory=lambda: {
            "attention_bias": False,
            "mlp_bias": False,
            "tie_word_embeddings": True,
        }
    )


@dataclass
class TrainingConfig:
    hyperparameters: Dict[str, Any] = field(
        default_factory=lambda: {"batch_size": 16, "learning_rate": 3e-4, "weight_decay": 0.01}
    )
    scheduler: Dict[str, Any] = field(
        default_factory=lambda: {"type": "cosine", "min_lr_ratio": 0.01}
    )
    optimization: Dict[str, Any] = field(default_factory=lambda: {"optimizer": "adamw"})
    epochs: Optional[int] = None
    early_stopping: Dict[str, Any] = field(
        default_factory=lambda: {
            "enabled": False,
            "patience": 3,
            "min_delta": 0.001,
            "metric": "val_loss",
            "mode": "min",
        }
    )
    lr_finder: Dict[str, Any] = field(
        default_factory=lambda: {
            "enabled": False,
            "min_lr": 1e-7,
            "max_lr": 1.0,
            "num_steps": 100,
        }
    )


@dataclass
class LoggingConfig:
    log_dir: str = "logs"
    checkpoint_dir: str = "checkpoints"
    steps: Dict[str, int] = field(
        default_factory=lambda: {
            "logging_interval": 1,
            "checkpoint_interval": 10000,
            "validation_interval": 0,
        }
    )
    metrics: Dict[str, bool] = field(
        default_factory=lambda: {
            "log_loss": True,
            "log_perplexity": True,
            "log_tokens_per_second": True,
            "log_learning_rate": True,
            "log_tokens_processed": True,
        }
    )
    tensorboard: bool = False
    wandb: bool = False
    wandb_project: Optional[str] = None
    wandb_entity: Optional[str] = None
    log_memory_usage: bool = False
    # per-step phase breakdown (data/compute/optimizer ms) appended to the
    # log line; measured with CUDA events, synchronized
    # only at logging steps
    log_step_breakdown: bool = False
    log_gradient_norm: bool = False
    log_parameter_norm: bool = False
    log_samples: bool = False
    log_samples_count: int = 3
    max_snapshots: int = 0  # >0: rotate old step checkpoints (reference train.py:79-80)
    # WebSocket stats mesh (reference stats_server/stats_client):
    # e.g. "ws://127.0.0.1:8765/ws" — each rank registers as rank<N> and
    # streams per-step metrics.
    stats_url: Optional[str] = None


@dataclass
class SystemConfig:
    seed: int = 42
    device: str = "gpu"  # "gpu" -> ROCm device, "cpu" -> CPU
    distributed: bool = False
    devices: Optional[List[str]] = None
    cuda_devices: Optional[List[int]] = None
    memory_limit: Optional[int] = None
    mixed_precision: bool = False
    precision: str = "bfloat16"  # float16 | bfloat16 | float32
    gradient_checkpointing: bool = False
    gradient_checkpointing_ratio: float = 0.5
    model_parallel: bool = False
    model_parallel_size: int = 1
    zero_optimization_level: int = 0
    # MI355X extensions
    distributed_backend: str = "nccl"  # "nccl" is RCCL on ROCm
    bucket_mb: int = 50
    # with model_parallel: S-shard the inter-sublayer activation stream
    # (Megatron sequence parallelism; parallel/tp.py)
    sequence_parallel: bool = False


@dataclass
class ResumeConfig:
    checkpoint: str
    reset_optimizer: bool = False
    reset_training_state: bool = False


def _filter_kwargs(cls, d: Dict[str, Any]) -> Dict[str, Any]:
    """Drop unknown keys so configs written for other versions still load."""
    names = {f.name for f in dataclasses.fields(cls)}
    return {k: v for k, v in d.items() if k in names}


@dataclass
class Config:
    name: str
    data: DataConfig = field(default_factory=DataConfig)
    model: ModelConfig = field(default_factory=ModelConfig)
    training: TrainingConfig = field(default_factory=TrainingConfig)
    logging: LoggingConfig = field(default_factory=LoggingConfig)
    system: SystemConfig = field(default_factory=SystemConfig)
    resume: Optional[ResumeConfig] = None
    overwrite: bool = False

    @classmethod
    def from_dict(cls, config_dict: Dict[str, Any]) -> "Config":
        if "name" not in config_dict:
            raise ValueError("Config must specify a 'name' field at the top level")
        training_dict = dict(config_dict.get("training", {}))
        epochs = training_dict.pop("epochs", None)
        resume = None
        if config_dict.get("resume"):
            resume = ResumeConfig(**_filter_kwargs(ResumeConfig, config_dict["resume"]))
        return cls(
            name=config_dict["name"],
            overwrite=config_dict.get("overwrite", False),
            data=DataConfig(**_filter_kwargs(DataConfig, config_dict.get("data", {}))),
            model=ModelConfig(**_filter_kwargs(ModelConfig, config_dict.get("model", {}))),
            training=TrainingConfig(
                **_filter_kwargs(TrainingConfig, training_dict), epochs=epochs
            ),
            logging=LoggingConfig(**_filter_kwargs(LoggingConfig, config_dict.get("logging", {}))),
            system=SystemConfig(**_filter_kwargs(SystemConfig, config_dict.get("system", {}))),
            resume=resume,
        )

    @classmethod
    def from_yaml(cls, yaml_path: str) -> "Config":
        config_dict = cls._load_yaml_with_extends(yaml_path)
        return cls.from_dict(config_dict)

    @staticmethod
    def _load_yaml_with_extends(yaml_path: str | Path, _depth: int = 0) -> Dict[str, Any]:
        """Load YAML with single-inheritance via ``extends: <path>`` (relative
        to the child file). The reference sketched a base-config scheme
        (/root/reference/configs/base_config.yaml:1-35) that no code path
        loaded; implemented for real here: child sections
        deep-merge over the base."""
        if _depth > 8:
            raise ValueError("config 'extends' chain too deep (cycle?)")
        with open(yaml_path, "r") as f:
            child = yaml.safe_load(f) or {}
        base_ref = child.pop("extends", None)
        if not base_ref:
            return child
        base_path = Path(yaml_path).parent / base_ref
        base = Config._load_yaml_with_extends(base_path, _depth + 1)

        def deep_merge(dst: Dict[str, Any], src: Dict[str, Any]) -> Dict[str, Any]:
            out = dict(dst)
            for k, v in src.items():
                if isinstance(v, dict) and isinstance(out.get(k), dict):
                    out[k] = deep_merge(out[k], v)
                else:
                    out[k] = v
            return out

        return deep_merge(base, child)

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        training = d["training"]
        epochs = training.pop("epochs", None)
        if epochs is not None:
            training["epochs"] = epochs
        if d.get("resume") is None:
            d.pop("resume", None)
        return d

    def save_yaml(self, path: str | Path) -> None:
        with open(path, "w") as f:
            yaml.safe_dump(self.to_dict(), f, sort_keys=False)
