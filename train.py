#!/usr/bin/env python3
"""Training entry point — keeps the reference's train.py surface
(/root/reference/train.py:174-176: `Trainer('cars_train_val').train()`) and
adds a real CLI (the reference has none — SURVEY.md §5.6).

Single GPU / CPU:
    python train.py --folder cars_train_val --model small --sidelength 64

8x MI355X data-parallel (RCCL over xGMI):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --model full --sidelength 128
"""

import argparse

from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig, load_yaml_config
from novel_view_synthesis_3d_amd.engine.trainer import Trainer


def main() -> None:
    ap = argparse.ArgumentParser(description="3DiM X-UNet trainer (MI355X)")
    ap.add_argument("--folder", default="cars_train_val",
                    help="SRN dataset root; synthetic data if missing")
    ap.add_argument("--config", default=None, help="YAML config file")
    ap.add_argument("--model", default="small",
                    help="model config name (tiny|small|full)")
    ap.add_argument("--batch-size", type=int, default=2)
    ap.add_argument("--lr", type=float, default=1e-4)
    ap.add_argument("--num-steps", type=int, default=100_000)
    ap.add_argument("--save-every", type=int, default=1000)
    ap.add_argument("--sidelength", type=int, default=64)
    ap.add_argument("--results-folder", default="./results")
    ap.add_argument("--loss", default="mse", choices=["mse", "frob"])
    ap.add_argument("--amp", default="bf16", choices=["bf16", "off"])
    ap.add_argument("--data", default="auto",
                    choices=["auto", "srn", "synthetic"])
    ap.add_argument("--resume", default=None,
                    help="checkpoint path, or 'auto'")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--profile", action="store_true",
                    help="torch.profiler trace of 5 steps, then exit")
    args = ap.parse_args()

    if args.config:
        cfgs = load_yaml_config(args.config)
        model_cfg, train_cfg = cfgs["model"], cfgs["train"]
    else:
        model_cfg = XUNetConfig.named(args.model)
        train_cfg = TrainConfig()
    train_cfg.loss = args.loss
    train_cfg.amp = args.amp
    train_cfg.data = args.data
    train_cfg.resume = args.resume
    train_cfg.seed = args.seed

    trainer = Trainer(args.folder,
                      train_batch_size=args.batch_size,
                      train_lr=args.lr,
                      train_num_steps=args.num_steps,
                      save_every=args.save_every,
                      img_sidelength=args.sidelength,
                      results_folder=args.results_folder,
                      model_cfg=model_cfg,
                      train_cfg=train_cfg)
    if args.profile:
        from novel_view_synthesis_3d_amd.engine.profiling import (
            profile_training,
        )
        profile_training(trainer)
        return
    trainer.train()


if __name__ == "__main__":
    main()
