#!/usr/bin/env python3
"""Sampling entry point — replaces the reference's import-time script
(/root/reference/sampling.py: hardcoded dirs, blocking cv2.imshow — D10).

    python sample.py --checkpoint checkpoints/model_00001000.pt \
        --folder cars_train_val --steps 256 --guidance 3 --out results/
"""

import argparse
import os

import numpy as np
import torch

from novel_view_synthesis_3d_amd.config import XUNetConfig
from novel_view_synthesis_3d_amd.data.srn import SceneClassDataset
from novel_view_synthesis_3d_amd.data.synthetic import synthetic_batch
from novel_view_synthesis_3d_amd.diffusion.sampler import DDPMSampler
from novel_view_synthesis_3d_amd.engine import checkpoint as ckpt
from novel_view_synthesis_3d_amd.models.xunet import XUNet


def main() -> None:
    ap = argparse.ArgumentParser(description="3DiM DDPM+CFG sampler (MI355X)")
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--folder", default=None,
                    help="SRN dataset root for conditioning views; synthetic "
                         "conditioning if omitted")
    ap.add_argument("--model", default=None,
                    help="model config name; default: read from checkpoint")
    ap.add_argument("--sidelength", type=int, default=None,
                    help="image sidelength; default: read from checkpoint")
    ap.add_argument("--batch-size", type=int, default=1)
    ap.add_argument("--steps", type=int, default=1000)
    ap.add_argument("--guidance", type=float, default=3.0)
    ap.add_argument("--out", default="./results")
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    # the checkpoint's extra payload records the model config and sidelength;
    # CLI flags override, and are required only for configless checkpoints
    payload = torch.load(args.checkpoint, map_location="cpu",
                         weights_only=False)
    extra = payload.get("extra", {}) if isinstance(payload, dict) else {}
    if args.model is not None:
        model_cfg = XUNetConfig.named(args.model)
    elif extra.get("model_cfg"):
        model_cfg = XUNetConfig(**extra["model_cfg"])
    else:
        model_cfg = XUNetConfig.named("small")
    sidelength = (args.sidelength if args.sidelength is not None
                  else int(extra.get("img_sidelength", 64)))
    args.sidelength = sidelength
    model = XUNet(model_cfg, sidelength).to(device)
    ckpt.load_checkpoint(args.checkpoint, model, map_location=device)
    model.eval()

    if args.folder and os.path.isdir(args.folder):
        ds = SceneClassDataset(root_dir=args.folder, max_num_instances=-1,
                               max_observations_per_instance=50,
                               img_sidelength=args.sidelength,
                               samples_per_instance=1)
        dl = torch.utils.data.DataLoader(ds, batch_size=args.batch_size,
                                         shuffle=True, drop_last=True,
                                         collate_fn=ds.collate_fn)
        raw, _ = next(iter(dl))
        cond = {k: v.to(device) for k, v in raw.items()
                if k in ("x", "R1", "t1", "R2", "t2", "K")}
    else:
        g = torch.Generator(device=device).manual_seed(args.seed)
        cond = synthetic_batch(args.batch_size, args.sidelength, device, g)
        cond.pop("x_target")

    sampler = DDPMSampler(model, num_steps=args.steps,
                          guidance_weight=args.guidance,
                          use_graph=(device == "cuda" and not args.no_graph))
    out = sampler.sample(cond)

    os.makedirs(args.out, exist_ok=True)
    img = ((out.clamp(-1, 1) * 0.5 + 0.5) * 255).to(torch.uint8).cpu().numpy()
    for i in range(img.shape[0]):
        path = os.path.join(args.out, f"sample_{i:03d}.png")
        try:
            from PIL import Image
            Image.fromarray(img[i]).save(path)
        except ImportError:
            np.save(path.replace(".png", ".npy"), img[i])
        print(f"wrote {path}")


if __name__ == "__main__":
    main()
