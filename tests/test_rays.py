"""Camera ray generation (K13) — geometric property tests."""

import math

import torch

from novel_view_synthesis_3d_amd.models.rays import camera_rays
from novel_view_synthesis_3d_amd.data.synthetic import random_cameras


def test_rays_shapes_and_unit_norm():
    g = torch.Generator().manual_seed(0)
    R, t, K = random_cameras(3, 16, generator=g)
    pos, d = camera_rays(R, t, K, 16, 16)
    assert pos.shape == (3, 16, 16, 3) and d.shape == (3, 16, 16, 3)
    assert torch.allclose(d.norm(dim=-1), torch.ones(3, 16, 16), atol=1e-5)
    # origins constant per image and equal to t
    assert torch.allclose(pos, t[:, None, None, :].expand_as(pos))


def test_principal_ray_is_optical_axis():
    """The ray through the principal point (cx, cy) must be the camera +z
    axis in world frame (third column of R)."""
    g = torch.Generator().manual_seed(1)
    R, t, K = random_cameras(2, 64, generator=g)
    H = 64
    pos, d = camera_rays(R, t, K, H, H)
    # principal point is at pixel center (H/2 - 0.5 + 0.5) = H/2 -> index 31.5;
    # use a camera with cx=cy=H/2: pixel (31,31) center is (31.5,31.5) != 32.
    # Instead check via direct projection: dir of pixel (v,u) must satisfy
    # K @ R^T d  proportional to (u+.5, v+.5, 1)
    u, v = 10, 37
    d_cam = torch.einsum("bij,bj->bi", R.transpose(1, 2), d[:, v, u])
    proj = torch.einsum("bij,bj->bi", K, d_cam)
    proj = proj / proj[:, 2:3]
    expect = torch.tensor([u + 0.5, v + 0.5, 1.0]).expand(2, 3)
    assert torch.allclose(proj, expect, atol=1e-3)


def test_look_at_points_toward_origin():
    g = torch.Generator().manual_seed(2)
    R, t, K = random_cameras(4, 32, generator=g)
    pos, d = camera_rays(R, t, K, 32, 32)
    # center-ish pixel direction should point roughly from eye toward origin
    center_dir = d[:, 16, 16]
    toward = -t / t.norm(dim=-1, keepdim=True)
    cos = (center_dir * toward).sum(-1)
    assert (cos > 0.98).all()
