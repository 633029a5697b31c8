"""Forward noising q(z_t | x0) — kernel K19 of SURVEY.md §2.4.

The reference computes this per-sample on the CPU inside dataset workers
(/root/reference/dataset/data_loader.py:99-100):

    z = sqrt(abar_t) * x0 + sqrt(1 - abar_t) * eps,   t ~ U[0, 1000)

Here it is a batched on-device op (plain torch tensor ops: a coefficient
gather + one fused-multiply-add over the (B,H,W,3) image batch — at 3
channels this is a few microseconds per step, so it is deliberately NOT a
hand-written kernel; rocprof shows it nowhere near the top-30 hotlist), so
dataset workers only move pixels.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from novel_view_synthesis_3d_amd.diffusion.schedules import (
    DiffusionSchedule, logsnr_schedule_cosine,
)


def q_sample(x0: torch.Tensor, t: torch.Tensor, schedule: DiffusionSchedule,
             noise: Optional[torch.Tensor] = None,
             generator: Optional[torch.Generator] = None,
             ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Noise a clean image batch to timestep t.

    Args:
      x0: (B, H, W, C) clean images in [-1, 1].
      t: (B,) int64 timesteps in [0, T).
    Returns: (z_t, noise, logsnr) — z_t same shape as x0, logsnr (B,) float32
    (the dataset attaches logsnr(t/T), reference data_loader.py:94-110).
    """
    if noise is None:
        noise = torch.empty_like(x0).normal_(generator=generator)
    tab = schedule.tables(x0.device, x0.dtype)
    shape = (t.shape[0],) + (1,) * (x0.ndim - 1)
    c1 = tab["sqrt_alphas_cumprod"].gather(0, t).reshape(shape)
    c2 = tab["sqrt_one_minus_alphas_cumprod"].gather(0, t).reshape(shape)
    z = c1 * x0 + c2 * noise
    logsnr = logsnr_schedule_cosine(
        t.to(torch.float32) / schedule.timesteps).to(torch.float32)
    return z, noise, logsnr
