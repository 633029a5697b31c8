"""Synthetic SRN-shaped data (no network, no datasets on disk).

Two forms:
* `synthetic_batch` — fabricate a full model-input batch directly on device
  (the bench path; modeled on reference train.py:23-34 `create_sample_data`,
  but with geometrically plausible cameras instead of uniform-random R/K).
* `SyntheticSceneDataset` — a torch Dataset with the same sample schema as
  data/srn.py, for trainer tests without files.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import numpy as np
import torch


_UP_CACHE = {}


def _look_at(eye: torch.Tensor) -> torch.Tensor:
    """cam->world rotation for a camera at `eye` looking at the origin.
    Camera convention: +z forward, +x right, +y down (pinhole with K as in
    data/io.py parse_intrinsics)."""
    fwd = -eye / eye.norm(dim=-1, keepdim=True).clamp_min(1e-8)
    key = str(eye.device)
    if key not in _UP_CACHE:
        _UP_CACHE[key] = torch.tensor([0.0, 0.0, 1.0], device=eye.device)
    up = _UP_CACHE[key].expand_as(fwd)
    right = torch.cross(fwd, up, dim=-1)
    right = right / right.norm(dim=-1, keepdim=True).clamp_min(1e-8)
    down = torch.cross(fwd, right, dim=-1)
    return torch.stack([right, down, fwd], dim=-1)  # columns = cam axes


_K_CACHE = {}


def _intrinsics(H: int, device) -> torch.Tensor:
    """Cached per-(device,H) so the bench's in-graph batch generation stays
    hipGraph-capturable (no H2D copies inside capture)."""
    key = (str(device), H)
    if key not in _K_CACHE:
        f = 1.75 * H
        _K_CACHE[key] = torch.tensor(
            [[f, 0.0, H / 2], [0.0, f, H / 2], [0.0, 0.0, 1.0]],
            device=device)
    return _K_CACHE[key]


def random_cameras(B: int, H: int, device="cpu",
                   generator: Optional[torch.Generator] = None):
    """Random cameras on a radius-1.3 sphere looking at the origin, with
    SRN-Cars-like intrinsics (f ~ 1.75*H)."""
    theta = torch.rand(B, device=device, generator=generator) * 2 * math.pi
    phi = torch.acos(
        0.9 * (2 * torch.rand(B, device=device, generator=generator) - 1))
    r = 1.3
    eye = torch.stack([r * torch.sin(phi) * torch.cos(theta),
                       r * torch.sin(phi) * torch.sin(theta),
                       r * torch.cos(phi)], dim=-1)
    R = _look_at(eye)
    K = _intrinsics(H, device).expand(B, 3, 3).contiguous()
    return R, eye, K


def synthetic_batch(B: int, H: int, device="cpu",
                    generator: Optional[torch.Generator] = None,
                    dtype: torch.dtype = torch.float32) -> Dict[str, torch.Tensor]:
    """Full model-input batch of the reference schema (train.py:53-60):
    x, x_target in [-1,1], plausible R1/t1/R2/t2/K. The trainer adds
    z/noise/logsnr via on-device q_sample (K19)."""
    dev = torch.device(device)
    x = torch.rand(B, H, H, 3, device=dev, generator=generator,
                   dtype=dtype) * 2 - 1
    x_target = torch.rand(B, H, H, 3, device=dev, generator=generator,
                          dtype=dtype) * 2 - 1
    R1, t1, K = random_cameras(B, H, dev, generator)
    R2, t2, _ = random_cameras(B, H, dev, generator)
    return {"x": x, "x_target": x_target,
            "R1": R1.to(dtype), "t1": t1.to(dtype),
            "R2": R2.to(dtype), "t2": t2.to(dtype), "K": K.to(dtype)}


class SyntheticSceneDataset(torch.utils.data.Dataset):
    """Deterministic synthetic scenes with the data/srn.py sample schema."""

    def __init__(self, num_instances: int = 4, views_per_instance: int = 8,
                 img_sidelength: int = 32, seed: int = 0):
        self.num_instances = num_instances
        self.views = views_per_instance
        self.H = img_sidelength
        self.seed = seed

    def __len__(self) -> int:
        return self.num_instances * self.views

    def _view(self, inst: int, view: int):
        g = torch.Generator().manual_seed(
            self.seed * 1_000_003 + inst * 1009 + view)
        img = torch.rand(self.H, self.H, 3, generator=g) * 2 - 1
        R, t, K = random_cameras(1, self.H, "cpu", g)
        return img, R[0], t[0], K[0]

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        inst, view = divmod(idx, self.views)
        view2 = int(torch.randint(self.views, (1,)))
        x, R1, t1, K = self._view(inst, view)
        xt, R2, t2, _ = self._view(inst, view2)
        return {"x": x, "x_target": xt, "R1": R1, "t1": t1,
                "R2": R2, "t2": t2, "K": K}

    @staticmethod
    def collate_fn(samples):
        return {k: torch.stack([s[k] for s in samples]) for k in samples[0]}
