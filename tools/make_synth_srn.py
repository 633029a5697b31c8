#!/usr/bin/env python3
"""Generate a synthetic SRN-layout PNG dataset (no network on the boxes, so
real ShapeNet renders are unavailable; this produces the same on-disk layout
the reference consumes — /root/reference/dataset/data_loader.py:40-50 — with
pose-consistent renders of a colored ball per instance, so a trained model
has real structure to denoise).

    python tools/make_synth_srn.py OUTDIR --instances 20 --views 25 --size 128
"""

import argparse
import os

import numpy as np


def look_at_pose(eye):
    """cam2world 4x4 looking at the origin (y-up)."""
    eye = np.asarray(eye, dtype=np.float64)
    fwd = -eye / np.linalg.norm(eye)           # camera looks along +z at obj
    up = np.array([0.0, 1.0, 0.0])
    right = np.cross(up, fwd)
    right /= np.linalg.norm(right) + 1e-12
    true_up = np.cross(fwd, right)
    pose = np.eye(4)
    pose[:3, 0] = right
    pose[:3, 1] = true_up
    pose[:3, 2] = fwd
    pose[:3, 3] = eye
    return pose


def render_ball(pose, f, cx, cy, size, color, radius=0.45):
    """Project a lambertian ball at the origin through a pinhole camera."""
    ys, xs = np.mgrid[0:size, 0:size].astype(np.float64)
    # camera ray dirs in world frame
    d_cam = np.stack([(xs - cx) / f, (ys - cy) / f, np.ones_like(xs)], -1)
    R = pose[:3, :3]
    d = d_cam @ R.T
    d /= np.linalg.norm(d, axis=-1, keepdims=True)
    o = pose[:3, 3]
    # ray-sphere: |o + t d|^2 = r^2
    b = 2.0 * d @ o
    c = o @ o - radius * radius
    disc = b * b - 4 * c
    hit = disc > 0
    t = np.where(hit, (-b - np.sqrt(np.maximum(disc, 0.0))) / 2.0, 0.0)
    p = o[None, None] + t[..., None] * d
    n = p / (np.linalg.norm(p, axis=-1, keepdims=True) + 1e-9)
    light = np.array([0.5, 0.8, 0.3])
    light = light / np.linalg.norm(light)
    lam = np.clip(n @ light, 0.1, 1.0)
    img = np.full((size, size, 3), 255, dtype=np.uint8)  # white background
    shade = (color[None, None] * lam[..., None] * 255).astype(np.uint8)
    img[hit] = shade[hit]
    return img


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("out")
    ap.add_argument("--instances", type=int, default=20)
    ap.add_argument("--views", type=int, default=25)
    ap.add_argument("--size", type=int, default=128)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    from PIL import Image

    rng = np.random.default_rng(args.seed)
    size = args.size
    f = 1.4 * size
    cx = cy = size / 2.0
    for i in range(args.instances):
        inst = os.path.join(args.out, f"inst_{i:04d}")
        os.makedirs(os.path.join(inst, "rgb"), exist_ok=True)
        os.makedirs(os.path.join(inst, "pose"), exist_ok=True)
        with open(os.path.join(inst, "intrinsics.txt"), "w") as fh:
            fh.write(f"{f} {cx} {cy} 0.\n0. 0. 0.\n1.\n{size} {size}\n")
        color = rng.uniform(0.2, 1.0, size=3)
        for v in range(args.views):
            theta = rng.uniform(0, 2 * np.pi)
            phi = rng.uniform(0.2, 1.3)
            r = 1.6
            eye = [r * np.cos(theta) * np.sin(phi), r * np.cos(phi),
                   r * np.sin(theta) * np.sin(phi)]
            pose = look_at_pose(eye)
            img = render_ball(pose, f, cx, cy, size, color)
            Image.fromarray(img).save(
                os.path.join(inst, "rgb", f"{v:06d}.png"))
            with open(os.path.join(inst, "pose", f"{v:06d}.txt"), "w") as fh:
                fh.write(" ".join(f"{x:.8f}" for x in pose.reshape(-1)))
    print(f"wrote {args.instances} instances x {args.views} views "
          f"({size}x{size}) under {args.out}")


if __name__ == "__main__":
    main()
