#!/usr/bin/env python3
"""Sampling benchmark — BASELINE.json config 5: DDPM 256-step reverse
sampling, full 128x128 model, CFG w=3, hipGraph-captured step.

    python tools/sample_bench.py --steps 256 --batch 64
"""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")

from novel_view_synthesis_3d_amd.config import XUNetConfig
from novel_view_synthesis_3d_amd.data.synthetic import synthetic_batch
from novel_view_synthesis_3d_amd.diffusion.sampler import DDPMSampler
from novel_view_synthesis_3d_amd.models.xunet import XUNet


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="full")
    ap.add_argument("--sidelength", type=int, default=128)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--steps", type=int, default=256)
    ap.add_argument("--no-graph", action="store_true")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    torch.manual_seed(0)
    model = XUNet(XUNetConfig.named(args.model), args.sidelength).cuda()
    model.eval()
    g = torch.Generator(device="cuda").manual_seed(0)
    cond = synthetic_batch(args.batch, args.sidelength, "cuda", g)
    cond.pop("x_target")

    sampler = DDPMSampler(model, num_steps=args.steps, guidance_weight=3.0,
                          use_graph=not args.no_graph)
    # warmup (includes capture)
    out = sampler.sample(cond)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = sampler.sample(cond)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert torch.isfinite(out).all()
    print(json.dumps({
        "metric": "DDPM 256-step sampling samples/sec (1 GPU)",
        "value": round(args.batch / dt, 3),
        "sec_per_batch": round(dt, 2),
        "ms_per_step": round(dt / args.steps * 1000, 2),
        "batch": args.batch, "steps": args.steps,
        "graph": not args.no_graph,
        "model": args.model, "sidelength": args.sidelength,
        "peak_mem_gb": round(torch.cuda.max_memory_allocated() / 1e9, 2),
    }), flush=True)


if __name__ == "__main__":
    main()
