"""Run-directory layout and checkpoint IO.

Keeps the reference's on-disk contract (/root/reference/core/training.py:169-195,
:1347-1394): ``runs/<name>/{log.txt, config.yaml, metadata.json, tokenizer/,
checkpoints/step_<N>_{model,optimizer}.safetensors + step_<N>_state.json}``.

Rank semantics for distributed runs (new here): only rank 0 writes; callers
barrier around save/load.
"""
from __future__ import annotations

import json
import re
import time
from pathlib import Path
from typing import Any, Dict, Optional, Tuple

import torch

try:
    from safetensors.torch import load_file as _st_load, save_file as _st_save

    _HAVE_SAFETENSORS = True
except Exception:  # pragma: no cover
    _HAVE_SAFETENSORS = False


class CheckpointManager:
    """Static helpers for the runs/ directory layout."""

    @staticmethod
    def validate_unique_name(name: str, runs_root: str | Path = "runs") -> None:
        run_path = Path(runs_root) / name
        if run_path.exists():
            raise ValueError(f"Run directory already exists for name '{name}'")

    @staticmethod
    def setup_run_directory(name: str, runs_root: str | Path = "runs") -> Tuple[Path, Path, Path]:
        run_dir = Path(runs_root) / name
        checkpoint_dir = run_dir / "checkpoints"
        run_dir.mkdir(parents=True, exist_ok=True)
        checkpoint_dir.mkdir(exist_ok=True)
        return run_dir, run_dir / "log.txt", checkpoint_dir

    @staticmethod
    def get_checkpoint_paths(checkpoint_path: str) -> Tuple[str, str, str]:
        return (
            f"{checkpoint_path}_model.safetensors",
            f"{checkpoint_path}_optimizer.safetensors",
            f"{checkpoint_path}_state.json",
        )


def _flatten_state(prefix: str, obj: Any, out: Dict[str, torch.Tensor]) -> Dict[str, Any]:
    """Flatten a nested optimizer-state structure into safetensors-compatible
    tensors plus a JSON-serializable skeleton describing non-tensor leaves."""
    if isinstance(obj, torch.Tensor):
        out[prefix] = obj.detach().contiguous().cpu()
        return {"__tensor__": prefix}
    if isinstance(obj, dict):
        return {str(k): _flatten_state(f"{prefix}.{k}", v, out) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_flatten_state(f"{prefix}.{i}", v, out) for i, v in enumerate(obj)]
    return {"__value__": obj}


def _unflatten_state(skel: Any, tensors: Dict[str, torch.Tensor]) -> Any:
    if isinstance(skel, dict):
        if "__tensor__" in skel:
            return tensors[skel["__tensor__"]]
        if "__value__" in skel:
            return skel["__value__"]
        out = {}
        for k, v in skel.items():
            kk: Any = k
            if re.fullmatch(r"-?\d+", k):
                kk = int(k)
            out[kk] = _unflatten_state(v, tensors)
        return out
    if isinstance(skel, list):
        return [_unflatten_state(v, tensors) for v in skel]
    return skel


def save_checkpoint(
    checkpoint_base: str,
    model: torch.nn.Module,
    optimizer_state: Optional[Dict[str, Any]],
    training_state: Dict[str, Any],
) -> None:
    model_path, opt_path, state_path = CheckpointManager.get_checkpoint_paths(checkpoint_base)
    state_dict = {k: v.detach().contiguous().cpu() for k, v in model.state_dict().items()}
    if not _HAVE_SAFETENSORS:  # pragma: no cover
        torch.save(state_dict, model_path)
    else:
        _st_save(state_dict, model_path)

    if optimizer_state is not None:
        tensors: Dict[str, torch.Tensor] = {}
        skeleton = _flatten_state("opt", optimizer_state, tensors)
        if _HAVE_SAFETENSORS:
            # safetensors needs at least the skeleton; store it in metadata
            _st_save(
                tensors if tensors else {"__empty__": torch.zeros(1)},
                opt_path,
                metadata={"skeleton": json.dumps(skeleton)},
            )
        else:  # pragma: no cover
            torch.save({"skeleton": skeleton, "tensors": tensors}, opt_path)

    training_state = dict(training_state)
    training_state.setdefault("saved_at", time.time())
    with open(state_path, "w") as f:
        json.dump(training_state, f, indent=2)


def load_checkpoint(
    checkpoint_base: str,
    model: Optional[torch.nn.Module] = None,
    map_location: str = "cpu",
    strict: bool = False,
) -> Tuple[Optional[Dict[str, Any]], Dict[str, Any]]:
    """Load a checkpoint triple. Returns (optimizer_state or None, training_state)."""
    model_path, opt_path, state_path = CheckpointManager.get_checkpoint_paths(checkpoint_base)
    if model is not None:
        if _HAVE_SAFETENSORS:
            sd = _st_load(model_path, device=map_location)
        else:  # pragma: no cover
            sd = torch.load(model_path, map_location=map_location)
        # Non-strict load like the reference (models/llama.py:414-477): drop keys
        # that do not match this architecture instead of erroring.
        model_keys = set(model.state_dict().keys())
        filtered = {k: v for k, v in sd.items() if k in model_keys}
        model.load_state_dict(filtered, strict=strict)

    optimizer_state = None
    if Path(opt_path).exists():
        if _HAVE_SAFETENSORS:
            import safetensors

            with safetensors.safe_open(opt_path, framework="pt", device=map_location) as f:
                meta = f.metadata() or {}
                tensors = {k: f.get_tensor(k) for k in f.keys()}
            skeleton = json.loads(meta.get("skeleton", "{}"))
            optimizer_state = _unflatten_state(skeleton, tensors)
        else:  # pragma: no cover
            blob = torch.load(opt_path, map_location=map_location)
            optimizer_state = _unflatten_state(blob["skeleton"], blob["tensors"])

    training_state: Dict[str, Any] = {}
    if Path(state_path).exists():
        with open(state_path) as f:
            training_state = json.load(f)
    return optimizer_state, training_state


def update_metadata(run_dir: Path, entry: Dict[str, Any]) -> None:
    """Append a checkpoint registry entry to runs/<name>/metadata.json
    (reference core/training.py:1369-1394)."""
    meta_path = run_dir / "metadata.json"
    meta: Dict[str, Any] = {}
    if meta_path.exists():
        try:
            meta = json.loads(meta_path.read_text())
        except json.JSONDecodeError:
            meta = {}
    meta.setdefault("checkpoints", [])
    meta["checkpoints"].append(entry)
    meta["updated_at"] = time.time()
    meta_path.write_text(json.dumps(meta, indent=2))


def latest_checkpoint(run_dir: str | Path) -> str | None:
    """Base path (``.../checkpoints/step_<N>`` or ``step_emergency_<N>``) of
    the run's newest checkpoint, or None. Used by ``--auto-resume`` (elastic
    restarts). Emergency checkpoints (written on training-loop failure)
    count too — after a crash they ARE the newest state; at equal step the
    regular snapshot wins. 'final' is ignored (a finished run has nothing
    to resume)."""
    ckdir = Path(run_dir) / "checkpoints"
    if not ckdir.is_dir():
        return None
    best = (-1, 0)  # (step, regular-beats-emergency)
    best_name = None
    for pth in ckdir.glob("step_*_state.json"):
        m = re.match(r"step_(emergency_)?(\d+)_state\.json", pth.name)
        if m:
            key = (int(m.group(2)), 0 if m.group(1) else 1)
            if key > best:
                best = key
                best_name = pth.name[: -len("_state.json")]
    if best_name is None:
        return None
    return str(ckdir / best_name)


def rotate_snapshots(checkpoint_dir: Path, max_snapshots: int) -> None:
    """Keep only the newest ``max_snapshots`` step checkpoints (never 'final').

    Parity with the older reference train.py's max_snapshots rotation
    (/root/reference/train.py:79-80).
    """
    if max_snapshots <= 0:
        return
    steps = []
    for p in checkpoint_dir.glob("step_*_state.json"):
        m = re.match(r"step_(\d+)_state\.json", p.name)
        if m:
            steps.append(int(m.group(1)))
    steps.sort()
    for s in steps[:-max_snapshots] if len(steps) > max_snapshots else []:
        for suffix in ("model.safetensors", "optimizer.safetensors", "state.json"):
            f = checkpoint_dir / f"step_{s}_{suffix}"
            if f.exists():
                f.unlink()
