"""CLI: ``python -m agilerl_amd.train <manifest.yaml> [--device ...]``.

Reference parity: ``agilerl/train.py:30-60``.
"""

from __future__ import annotations

import argparse

import torch


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="agilerl-amd manifest trainer")
    p.add_argument("manifest", help="path to a training manifest YAML")
    p.add_argument("--device", default=None, help="cpu / cuda:N (default: auto)")
    p.add_argument("--checkpoint-steps", type=int, default=None)
    p.add_argument("--checkpoint-path", default=None)
    p.add_argument("--wb", action="store_true", help="log to wandb if available")
    p.add_argument("--csv", default=None, help="CSV log path")
    p.add_argument("--tensorboard", default=None, help="TensorBoard log dir")
    p.add_argument("--prometheus-port", type=int, default=None,
                   help="export population metrics as Prometheus gauges")
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")

    from .logger import make_loggers
    from .training.trainer import LocalTrainer

    loggers = make_loggers(
        stdout=True,
        csv_path=args.csv,
        tensorboard_dir=args.tensorboard,
        wandb_project="agilerl-amd" if args.wb else None,
        prometheus_port=args.prometheus_port,
    )
    trainer = LocalTrainer.from_manifest(args.manifest, device=device, loggers=loggers)
    if args.checkpoint_steps is not None:
        trainer.manifest.training.checkpoint = args.checkpoint_steps
    if args.checkpoint_path is not None:
        trainer.manifest.training.checkpoint_path = args.checkpoint_path
    return trainer.train()


if __name__ == "__main__":
    main()
