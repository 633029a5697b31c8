"""Per-pixel pinhole camera rays — kernel K13 of SURVEY.md §2.4.

Replaces the reference's visu3d dependency (/root/reference/model/xunet.py:159-171):

    world_from_cam = v3d.Transform(R=batch['R1'], t=batch['t1'])
    rays = v3d.Camera(spec=v3d.PinholeCamera(resolution=(H,W), K=batch['K']),
                      world_from_cam=world_from_cam).rays()

Convention (visu3d semantics): `R` is the cam->world rotation, `t` the camera
origin in world coordinates; a pixel at (row v, col u) has its center at
(u + 0.5, v + 0.5); the camera-frame direction is K^{-1} [u+.5, v+.5, 1]^T;
ray.pos = t (constant per image), ray.dir = normalize(R @ dir_cam).
"""

from __future__ import annotations

import torch


def inv3x3(K: torch.Tensor) -> torch.Tensor:
    """Closed-form batched 3x3 inverse (adjugate/det). Unlike
    torch.linalg.inv this is pure elementwise math: hipGraph-capture-safe
    and faster for the tiny camera matrices."""
    K = K.to(torch.float32)
    a, b, c = K[..., 0, 0], K[..., 0, 1], K[..., 0, 2]
    d, e, f = K[..., 1, 0], K[..., 1, 1], K[..., 1, 2]
    g, h, i = K[..., 2, 0], K[..., 2, 1], K[..., 2, 2]
    A = e * i - f * h
    B = c * h - b * i
    C = b * f - c * e
    D = f * g - d * i
    E = a * i - c * g
    F = c * d - a * f
    G = d * h - e * g
    Hc = b * g - a * h
    I = a * e - b * d
    det = a * A + b * D + c * G
    inv = torch.stack([
        torch.stack([A, B, C], dim=-1),
        torch.stack([D, E, F], dim=-1),
        torch.stack([G, Hc, I], dim=-1)], dim=-2)
    return inv / det[..., None, None]


def camera_rays(R: torch.Tensor, t: torch.Tensor, K: torch.Tensor,
                H: int, W: int):
    """Compute per-pixel ray origins and unit directions.

    Args:
      R: (B, 3, 3) cam->world rotation.
      t: (B, 3) camera origin in world frame.
      K: (B, 3, 3) pinhole intrinsics [[fx,0,cx],[0,fy,cy],[0,0,1]].
    Returns:
      pos: (B, H, W, 3) ray origins (t broadcast over pixels).
      dir: (B, H, W, 3) unit ray directions in world frame.
    """
    B = R.shape[0]
    device, dtype = R.device, R.dtype
    v, u = torch.meshgrid(
        torch.arange(H, device=device, dtype=dtype) + 0.5,
        torch.arange(W, device=device, dtype=dtype) + 0.5,
        indexing="ij")
    px = torch.stack([u, v, torch.ones_like(u)], dim=-1)        # (H, W, 3)
    Kinv = inv3x3(K).to(dtype)                                  # (B, 3, 3)
    # dir_cam[b,h,w,:] = Kinv[b] @ px[h,w,:]
    dir_cam = torch.einsum("bij,hwj->bhwi", Kinv, px)
    dir_world = torch.einsum("bij,bhwj->bhwi", R, dir_cam)
    dir_world = dir_world / dir_world.norm(dim=-1, keepdim=True).clamp_min(1e-12)
    pos = t[:, None, None, :].expand(B, H, W, 3)
    return pos, dir_world
