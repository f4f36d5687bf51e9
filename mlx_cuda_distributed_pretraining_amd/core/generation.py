"""CLI: generate text from a trained run.

Parity with /root/reference/core/generation.py:10-81: loads the run's config
and final checkpoint, builds samplers, generates.
"""
from __future__ import annotations

import argparse
from pathlib import Path

from .config import Config
from .trainer import Trainer
from ..inference.generate import beam_search, generate


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Generate from a trained run")
    p.add_argument("--run", type=str, required=True, help="run name under runs/")
    p.add_argument("--prompt", type=str, default="")
    p.add_argument("--max-tokens", type=int, default=128)
    p.add_argument("--temperature", type=float, default=0.0)
    p.add_argument("--top-p", type=float, default=1.0)
    p.add_argument("--min-p", type=float, default=0.0)
    p.add_argument("--repetition-penalty", type=float, default=None)
    p.add_argument("--beam", type=int, default=0, help=">0: beam search with this width")
    p.add_argument("--checkpoint", type=str, default="final")
    p.add_argument("--runs-root", type=str, default="runs")
    args = p.parse_args(argv)

    run_dir = Path(args.runs_root) / args.run
    config = Config.from_yaml(str(run_dir / "config.yaml"))
    config.overwrite = True
    trainer = Trainer(config, for_training=False, runs_root=args.runs_root)
    ckpt = run_dir / "checkpoints" / f"step_{args.checkpoint}_model.safetensors"
    trainer.model.load_weights(str(ckpt))

    if args.beam > 0:
        results = beam_search(
            trainer.model, trainer.tokenizer, args.prompt,
            max_tokens=args.max_tokens, beam_width=args.beam,
        )
        for text, score in results:
            print(f"[{score:.3f}] {text}")
    else:
        text, stats = generate(
            trainer.model, trainer.tokenizer, args.prompt,
            max_tokens=args.max_tokens, temperature=args.temperature,
            top_p=args.top_p, min_p=args.min_p,
            repetition_penalty=args.repetition_penalty, verbose=True,
        )
        print(text)


if __name__ == "__main__":
    main()
