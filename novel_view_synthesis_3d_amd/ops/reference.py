"""Eager PyTorch reference implementations of every fused op.

These are the numerics oracle (tests compare the HIP kernels against these in
fp32) and the CPU execution path. Shapes follow the framework's activation
layout: (B, F=2, H, W, C) contiguous — NHWC with a frame axis, the layout the
reference's FLAX model uses (/root/reference/model/xunet.py:228) and the
layout the CDNA4 kernels want for coalesced 64-lane access.

Kernel inventory: SURVEY.md §2.4 (K1-K20).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


SQRT_HALF = 1.0 / math.sqrt(2.0)


def _same_pad(size: int, k: int, s: int) -> Tuple[int, int]:
    """FLAX 'SAME' padding (asymmetric: low = total//2): matches nn.Conv."""
    out = -(-size // s)  # ceil
    total = max((out - 1) * s + k - size, 0)
    return total // 2, total - total // 2


def frame_conv3x3(x: torch.Tensor, weight: torch.Tensor,
                  bias: Optional[torch.Tensor], stride: int = 1) -> torch.Tensor:
    """K1/K2: per-frame 3x3 'SAME' conv over (B,F,H,W,Cin) -> (B,F,H',W',Cout).

    The reference's only conv type: kernel (1,3,3), stride (1,s,s)
    (/root/reference/model/xunet.py:81,85,199-202,229,276). `weight` is
    (Cout, 3, 3, Cin) contiguous (OHWI) — presented to conv2d as a
    channels-last (Cout,Cin,3,3) view so MIOpen takes the NHWC path with no
    transposes.
    """
    B, Fr, H, W, C = x.shape
    ph = _same_pad(H, 3, stride)
    pw = _same_pad(W, 3, stride)
    xf = x.reshape(B * Fr, H, W, C).permute(0, 3, 1, 2)  # NCHW view of NHWC data
    w = weight.permute(0, 3, 1, 2)  # channels-last view of OHWI storage
    if ph != (1, 1) or pw != (1, 1) or stride != 1:
        xf = F.pad(xf, (pw[0], pw[1], ph[0], ph[1]))
        y = F.conv2d(xf, w, bias, stride=stride)
    else:
        y = F.conv2d(xf, w, bias, stride=1, padding=1)
    Ho, Wo = y.shape[-2], y.shape[-1]
    return y.permute(0, 2, 3, 1).reshape(B, Fr, Ho, Wo, -1)


def joint_groupnorm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                    groups: int, eps: float = 1e-6,
                    film: Optional[torch.Tensor] = None,
                    silu: bool = False, p_drop: float = 0.0) -> torch.Tensor:
    """K3(+K5+K4): GroupNorm with statistics jointly over BOTH frames and all
    spatial positions per (batch, group) — the reference's frame-axis GroupNorm
    (/root/reference/model/xunet.py:46-52; flax GroupNorm reduces over all
    non-batch axes). Optionally fused FiLM modulate (xunet.py:54-61) and SiLU.

    x: (B, F, H, W, C); gamma/beta: (C,); film: (B, F, H, W, 2C) packed
    scale|shift (one Dense output, consumed strided — no split copies).
    Stats in fp32; output in x.dtype.
    """
    B, Fr, H, W, C = x.shape
    assert C % groups == 0, f"C={C} not divisible by groups={groups}"
    xf = x.to(torch.float32).reshape(B, Fr, H, W, groups, C // groups)
    mean = xf.mean(dim=(1, 2, 3, 5), keepdim=True)
    var = xf.var(dim=(1, 2, 3, 5), unbiased=False, keepdim=True)
    h = (xf - mean) * torch.rsqrt(var + eps)
    h = h.reshape(B, Fr, H, W, C)
    h = h * gamma.to(torch.float32) + beta.to(torch.float32)
    if film is not None:
        scale = film[..., :C].to(torch.float32)
        shift = film[..., C:].to(torch.float32)
        h = h * (1.0 + scale) + shift
    if silu:
        h = F.silu(h)
    if p_drop > 0.0:
        h = F.dropout(h, p_drop, training=True)
    return h.to(x.dtype)


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """K7: multi-head scaled dot-product attention over flattened H*W tokens.

    q/k/v: (B, L, h, d) -> (B, L, h, d). Matches flax nn.dot_product_attention
    (/root/reference/model/xunet.py:103): softmax(q k^T / sqrt(d)) v, softmax in
    fp32. NOTE the reference has NO output projection (xunet.py:126 commented
    out) — heads are just reshaped back by the caller.
    """
    B, L, h, d = q.shape
    qt = q.permute(0, 2, 1, 3)  # (B, h, L, d)
    kt = k.permute(0, 2, 1, 3)
    vt = v.permute(0, 2, 1, 3)
    scores = torch.matmul(qt, kt.transpose(-1, -2)) * (1.0 / math.sqrt(d))
    p = torch.softmax(scores.to(torch.float32), dim=-1).to(q.dtype)
    out = torch.matmul(p, vt)
    return out.permute(0, 2, 1, 3)


def nearest_upsample2x(x: torch.Tensor) -> torch.Tensor:
    """K8: nearest-neighbor 2x upsample (/root/reference/model/xunet.py:14-18)."""
    B, Fr, H, W, C = x.shape
    x = x.reshape(B, Fr, H, 1, W, 1, C).expand(B, Fr, H, 2, W, 2, C)
    return x.reshape(B, Fr, 2 * H, 2 * W, C)


def avgpool_downsample2x(x: torch.Tensor) -> torch.Tensor:
    """K9: 2x2 average-pool downsample (/root/reference/model/xunet.py:20-21)."""
    B, Fr, H, W, C = x.shape
    x = x.reshape(B, Fr, H // 2, 2, W // 2, 2, C)
    return x.mean(dim=(3, 5))


def residual_scale_add(h: torch.Tensor, h_in: torch.Tensor) -> torch.Tensor:
    """K10: (h + h_in) / sqrt(2) (/root/reference/model/xunet.py:92,127)."""
    return (h + h_in) * SQRT_HALF


def posenc_ddpm(timesteps: torch.Tensor, emb_ch: int,
                max_time: float = 1000.0) -> torch.Tensor:
    """K12 part: DDPM sinusoidal positional encoding
    (/root/reference/model/xunet.py:23-35). timesteps: (B,) -> (B, emb_ch)."""
    timesteps = timesteps.to(torch.float32) * (1000.0 / max_time)
    half = emb_ch // 2
    freq = torch.exp(
        torch.arange(half, dtype=torch.float32, device=timesteps.device)
        * (-math.log(10000.0) / (half - 1)))
    ang = timesteps[:, None] * freq[None, :]
    return torch.cat([torch.sin(ang), torch.cos(ang)], dim=-1)


def squash_logsnr(logsnr: torch.Tensor) -> torch.Tensor:
    """K12 part: clip(+-20) then 2*atan(e^{-l/2})/pi in [0,1]
    (/root/reference/model/xunet.py:152-153)."""
    l = torch.clamp(logsnr.to(torch.float32), -20.0, 20.0)
    return 2.0 * torch.atan(torch.exp(-l / 2.0)) / math.pi


def posenc_nerf(x: torch.Tensor, min_deg: int = 0, max_deg: int = 15) -> torch.Tensor:
    """K14: NeRF positional encoding (/root/reference/model/xunet.py:37-44).

    Concat [x, sin(x*2^i), sin(x*2^i + pi/2)] for i in [min_deg, max_deg).
    x: (..., D) -> (..., D*(1 + 2*(max_deg-min_deg))).
    """
    if min_deg == max_deg:
        return x
    scales = torch.tensor([2.0 ** i for i in range(min_deg, max_deg)],
                          dtype=x.dtype, device=x.device)
    xb = (x[..., None, :] * scales[:, None]).reshape(*x.shape[:-1], -1)
    return torch.cat([x, torch.sin(xb), torch.sin(xb + math.pi / 2.0)], dim=-1)


def pose_embedding(R: torch.Tensor, t: torch.Tensor, K: torch.Tensor,
                   cond_mask: Optional[torch.Tensor], H: int, W: int,
                   out_dtype: torch.dtype) -> torch.Tensor:
    """K13+K14+K15 fused: per-pixel rays for BOTH cameras -> NeRF posenc ->
    CFG mask. R: (B,2,3,3), t: (B,2,3), K: (B,3,3). Returns (B,2,H,W,144).

    Eager oracle for the rays_posenc HIP kernel
    (reference model/xunet.py:159-179)."""
    from novel_view_synthesis_3d_amd.models.rays import camera_rays
    embs = []
    for f in range(2):
        pos, direc = camera_rays(R[:, f], t[:, f], K, H, W)
        embs.append(torch.cat([posenc_nerf(pos, 0, 15),
                               posenc_nerf(direc, 0, 8)], dim=-1))
    pe = torch.stack(embs, dim=1)
    if cond_mask is not None:
        pe = pe * cond_mask.to(pe.dtype).reshape(-1, 1, 1, 1, 1)
    return pe.to(out_dtype)
