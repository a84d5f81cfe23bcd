"""CLI: ``python -m agilerl_amd.train <manifest.yaml> [--device ...]``.

Reference parity: ``agilerl/train.py:30-60``.
"""

from __future__ import annotations

import argparse

import torch


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="agilerl-amd manifest trainer")
    p.add_argument("manifest", help="path to a training manifest YAML")
    p.add_argument("-d", "--device", default=None, help="cpu / cuda:N (default: auto)")
    p.add_argument("--checkpoint-steps", type=int, default=None)
    p.add_argument("--checkpoint-path", default=None)
    p.add_argument("--overwrite-checkpoints", action="store_true")
    p.add_argument("--resume-from-checkpoint", default=None)
    p.add_argument("--save-elite", action="store_true",
                   help="persist the elite agent after training")
    p.add_argument("--elite-path", default=None)
    p.add_argument("--wb", action="store_true", help="log to wandb if available")
    p.add_argument("--wandb-api-key", default=None)
    p.add_argument("--csv", default=None, help="CSV log path")
    p.add_argument("--tensorboard", default=None, help="TensorBoard log dir")
    p.add_argument("--tensorboard-log-dir", default=None,
                   help="alias of --tensorboard (reference train.py flag)")
    p.add_argument("--verbose", action=argparse.BooleanOptionalAction, default=True)
    p.add_argument("--use-accelerator", action="store_true",
                   help="accepted for reference compatibility; distributed "
                        "training uses torchrun + agilerl_amd.parallel")
    p.add_argument("--prometheus-port", type=int, default=None,
                   help="export population metrics as Prometheus gauges")
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")

    from .logger import make_loggers
    from .training.trainer import LocalTrainer

    if args.use_accelerator:
        import warnings

        warnings.warn(
            "--use-accelerator is accepted for reference compatibility but "
            "ignored: launch with torchrun for multi-GPU (one process per "
            "GPU over RCCL).", RuntimeWarning,
        )
    if args.wandb_api_key:
        import os

        os.environ.setdefault("WANDB_API_KEY", args.wandb_api_key)
    loggers = make_loggers(
        stdout=bool(args.verbose),
        csv_path=args.csv,
        tensorboard_dir=args.tensorboard or args.tensorboard_log_dir,
        wandb_project="agilerl-amd" if args.wb else None,
        prometheus_port=args.prometheus_port,
    )
    trainer = LocalTrainer.from_manifest(args.manifest, device=device, loggers=loggers)
    t = trainer.manifest.training
    if args.checkpoint_steps is not None:
        t.checkpoint = args.checkpoint_steps
    if args.checkpoint_path is not None:
        t.checkpoint_path = args.checkpoint_path
    if args.overwrite_checkpoints:
        t.overwrite_checkpoints = True
    if args.resume_from_checkpoint is not None:
        t.resume_from_checkpoint = args.resume_from_checkpoint
    if args.save_elite:
        t.save_elite = True
    if args.elite_path is not None:
        t.elite_path = args.elite_path
    return trainer.train()


if __name__ == "__main__":
    main()
