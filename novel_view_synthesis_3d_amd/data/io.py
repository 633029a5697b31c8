"""SRN/ShapeNet-layout IO: images, poses, intrinsics.

Re-implements /root/reference/dataset/data_util.py:12-24,43-52,67-72,101-105
and /root/reference/dataset/util.py:46-81 without the cv2/imageio/skimage
dependencies (not in this image): PNG/JPG via PIL when available, .npy arrays
always; INTER_AREA resize via torch's area interpolation (identical for
integer downscale factors).
"""

from __future__ import annotations

import os
from glob import glob
from typing import Optional, Tuple

import numpy as np
import torch
import torch.nn.functional as F

IMG_EXTS = ("*.png", "*.jpg", "*.JPEG", "*.JPG", "*.npy")


def glob_imgs(path: str):
    imgs = []
    for ext in IMG_EXTS:
        imgs.extend(glob(os.path.join(path, ext)))
    return sorted(imgs)


def square_crop(img: np.ndarray) -> np.ndarray:
    """Center square crop (reference data_util.py:67-72)."""
    h, w = img.shape[:2]
    m = min(h, w)
    cy, cx = h // 2, w // 2
    return img[cy - m // 2:cy + m // 2, cx - m // 2:cx + m // 2]


def area_resize(img: np.ndarray, sidelength: int) -> np.ndarray:
    """cv2.INTER_AREA-equivalent resize (exact for integer downscale)."""
    if img.shape[0] == sidelength and img.shape[1] == sidelength:
        return img
    t = torch.from_numpy(np.ascontiguousarray(img)).permute(2, 0, 1)[None].float()
    if img.shape[0] % sidelength == 0:
        out = F.interpolate(t, size=(sidelength, sidelength), mode="area")
    else:  # upscale or non-integer: bilinear like cv2 does effectively
        out = F.interpolate(t, size=(sidelength, sidelength), mode="bilinear",
                            align_corners=False)
    return out[0].permute(1, 2, 0).numpy()


def load_rgb(path: str, sidelength: Optional[int] = None) -> np.ndarray:
    """Load an image to float32 HWC in [-1, 1] (reference data_util.py:12-24
    returns CHW; we keep HWC since the model layout is NHWC)."""
    if path.endswith(".npy"):
        img = np.load(path).astype(np.float32)
        if img.max() > 1.5:
            img = img / 255.0
    else:
        from PIL import Image
        with Image.open(path) as im:
            img = np.asarray(im.convert("RGB"), dtype=np.float32) / 255.0
    img = img[:, :, :3]
    img = square_crop(img)
    if sidelength is not None:
        img = area_resize(img, sidelength)
    return (img - 0.5) * 2.0


def load_pose(path: str) -> np.ndarray:
    """4x4 pose from txt, 1-line (16 floats) or 4-line format
    (reference data_util.py:43-52)."""
    with open(path) as f:
        lines = f.read().splitlines()
    if len(lines) == 1:
        vals = [float(x) for x in lines[0].split()]
        return np.asarray(vals, dtype=np.float32).reshape(4, 4)
    rows = [[float(v) for v in ln.split()[:4]] for ln in lines[:4]]
    return np.asarray(rows, dtype=np.float32)


def parse_intrinsics(filepath: str, trgt_sidelength: Optional[int] = None,
                     invert_y: bool = False
                     ) -> Tuple[np.ndarray, np.ndarray, float, bool]:
    """Parse SRN intrinsics.txt (reference util.py:46-81).

    Line 1: f cx cy _ ; line 2: grid barycenter (3); line 3: scale;
    line 4: height width ; line 5 (optional): world2cam flag.
    f/cx/cy are rescaled to trgt_sidelength. Returns (4x4 K, barycenter,
    scale, world2cam_poses).
    """
    with open(filepath) as f:
        fval, cx, cy, _ = map(float, f.readline().split())
        barycenter = np.array(list(map(float, f.readline().split())),
                              dtype=np.float32)
        scale = float(f.readline())
        height, width = map(float, f.readline().split())
        try:
            world2cam = bool(int(f.readline()))
        except (ValueError, TypeError):
            world2cam = False

    if trgt_sidelength is not None:
        cx = cx / width * trgt_sidelength
        cy = cy / height * trgt_sidelength
        fval = trgt_sidelength / height * fval

    fx = fval
    fy = -fval if invert_y else fval
    K = np.array([[fx, 0.0, cx, 0.0],
                  [0.0, fy, cy, 0.0],
                  [0.0, 0.0, 1.0, 0.0],
                  [0.0, 0.0, 0.0, 1.0]], dtype=np.float32)
    return K, barycenter, scale, world2cam
