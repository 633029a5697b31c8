#!/usr/bin/env python3
"""Flagship training benchmark — BASELINE.json north-star metric:
train images/sec (whole node), X-UNet 128x128 SRN-shaped synthetic data.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU. W untimed warmup
steps, then exactly K timed steps bracketed by barrier+synchronize on both
sides; elapsed is the MAX over ranks; rank 0 prints ONE JSON line.

The timed step is the COMPLETE training step: on-device synthetic batch
generation + forward noising, bf16 forward, loss, backward, bucketed RCCL
gradient all-reduce, fp32 Adam update.
"""

import argparse
import json
import os
import time

import torch

from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
from novel_view_synthesis_3d_amd.engine.trainer import Trainer


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=96,
                    help="per-GPU batch size (weak scaling)")
    ap.add_argument("--model", default="full")
    ap.add_argument("--sidelength", type=int, default=128)
    ap.add_argument("--amp", default="bf16", choices=["bf16", "off"])
    ap.add_argument("--graph", default="off", choices=["on", "off"],
                    help="hipGraph-capture the train step (single-node)")
    ap.add_argument("--bucket-mb", type=float, default=40.0,
                    help="DDP gradient bucket size (xGMI sweep hook: "
                         "try 25/40/80)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    n_gpus = world if world > 1 else args.gpus

    use_cuda = torch.cuda.is_available()
    model_cfg = XUNetConfig.named(args.model)
    train_cfg = TrainConfig()
    train_cfg.data = "synthetic"
    train_cfg.amp = args.amp
    train_cfg.seed = 1234
    train_cfg.use_graph = (args.graph == "on" and use_cuda)
    train_cfg.bucket_mb = args.bucket_mb

    trainer = Trainer(None,
                      train_batch_size=args.batch,
                      img_sidelength=args.sidelength,
                      train_num_steps=10 ** 9,
                      model_cfg=model_cfg,
                      train_cfg=train_cfg,
                      results_folder=os.path.join(
                          os.environ.get("TMPDIR", "/tmp"), "nvs3d_bench"))

    def barrier_sync():
        if trainer.ddp.enabled:
            torch.distributed.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        trainer.train_step()
    barrier_sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = trainer.train_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    elapsed = trainer.ddp.max_scalar(elapsed)  # MAX over ranks

    if rank == 0:
        global_batch = args.batch * n_gpus
        images_per_sec = global_batch * args.steps / elapsed
        result = {
            "metric": "train images/sec (whole node), X-UNet 128x128 SRN",
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16" if (args.amp == "bf16" and use_cuda) else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"X-UNet {args.model} ch={model_cfg.ch} "
                         f"ch_mult={list(model_cfg.ch_mult)}",
                "img_sidelength": args.sidelength,
                "global_batch": global_batch,
                "per_gpu_batch": args.batch,
                "parallelism": f"dp{n_gpus}",
                "bucket_mb": args.bucket_mb,
                "final_loss": float(loss.item()),
                "params": trainer.model.num_params(),
                "graph": getattr(trainer, "graph_active", False),
                "peak_mem_gb": round(
                    torch.cuda.max_memory_allocated() / 1e9, 2)
                if use_cuda else None,
            },
        }
        print(json.dumps(result), flush=True)

    if trainer.ddp.enabled:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
