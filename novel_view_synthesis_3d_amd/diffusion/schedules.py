"""Diffusion schedules — the single source of truth.

The reference duplicates this math in three places with two array libraries
(SURVEY.md D9): /root/reference/sampling.py:16-41,73-76,
/root/reference/dataset/data_loader.py:15-25,94-97. Here it lives once, in
torch, with float64 internal precision and cached per-device float32 tables.

Math (Nichol & Dhariwal cosine schedule, s=0.008, T=1000, beta clipped to
0.9999 — reference sampling.py:16-26):

    alpha_bar(t) = cos^2(((t/T + s)/(1+s)) * pi/2) / cos^2((s/(1+s)) * pi/2)
    beta_t       = 1 - alpha_bar(t)/alpha_bar(t-1)

Continuous cosine logsnr schedule (reference sampling.py:73-76):

    logsnr(u) = -2 log(tan(a*u + b)),  b = arctan(e^{-logsnr_max/2}),
                a = arctan(e^{-logsnr_min/2}) - b,  u in [0, 1]
"""

from __future__ import annotations

import math
from typing import Union

import torch

Number = Union[float, torch.Tensor]


def cosine_beta_schedule(timesteps: int = 1000, s: float = 0.008,
                         max_beta: float = 0.9999) -> torch.Tensor:
    """Cosine beta schedule; float64; matches reference sampling.py:16-26."""
    steps = timesteps + 1
    x = torch.linspace(0, timesteps, steps, dtype=torch.float64)
    alphas_cumprod = torch.cos(((x / timesteps) + s) / (1 + s) * math.pi * 0.5) ** 2
    alphas_cumprod = alphas_cumprod / alphas_cumprod[0]
    betas = 1 - (alphas_cumprod[1:] / alphas_cumprod[:-1])
    return torch.clip(betas, 0, max_beta)


def logsnr_schedule_cosine(t: Number, *, logsnr_min: float = -20.0,
                           logsnr_max: float = 20.0) -> Number:
    """Continuous cosine logsnr(t), t in [0,1]; matches reference sampling.py:73-76."""
    b = math.atan(math.exp(-0.5 * logsnr_max))
    a = math.atan(math.exp(-0.5 * logsnr_min)) - b
    if isinstance(t, torch.Tensor):
        return -2.0 * torch.log(torch.tan(a * t + b))
    return -2.0 * math.log(math.tan(a * t + b))


def t_from_logsnr(logsnr: Number, *, logsnr_min: float = -20.0,
                  logsnr_max: float = 20.0) -> Number:
    """Inverse of logsnr_schedule_cosine; matches reference sampling.py:120-123."""
    b = math.atan(math.exp(-0.5 * logsnr_max))
    a = math.atan(math.exp(-0.5 * logsnr_min)) - b
    if isinstance(logsnr, torch.Tensor):
        return (torch.atan(torch.exp(-0.5 * logsnr)) - b) / a
    return (math.atan(math.exp(-0.5 * logsnr)) - b) / a


class DiffusionSchedule:
    """Precomputed DDPM tables (reference sampling.py:28-41), float32, device-cached.

    Tables:
      betas, alphas_cumprod, sqrt_alphas_cumprod, sqrt_one_minus_alphas_cumprod,
      sqrt_recip_alphas_cumprod, sqrt_recipm1_alphas_cumprod,
      posterior_variance, posterior_log_variance_clipped,
      posterior_mean_coef1, posterior_mean_coef2
    """

    TABLE_NAMES = (
        "betas", "alphas_cumprod", "alphas_cumprod_prev",
        "sqrt_alphas_cumprod", "sqrt_one_minus_alphas_cumprod",
        "sqrt_recip_alphas_cumprod", "sqrt_recipm1_alphas_cumprod",
        "posterior_variance", "posterior_log_variance_clipped",
        "posterior_mean_coef1", "posterior_mean_coef2",
    )

    def __init__(self, timesteps: int = 1000, s: float = 0.008):
        self.timesteps = timesteps
        betas = cosine_beta_schedule(timesteps, s)
        alphas = 1.0 - betas
        alphas_cumprod = torch.cumprod(alphas, dim=0)
        alphas_cumprod_prev = torch.cat(
            [torch.ones(1, dtype=torch.float64), alphas_cumprod[:-1]])

        t = {}
        t["betas"] = betas
        t["alphas_cumprod"] = alphas_cumprod
        t["alphas_cumprod_prev"] = alphas_cumprod_prev
        t["sqrt_alphas_cumprod"] = torch.sqrt(alphas_cumprod)
        t["sqrt_one_minus_alphas_cumprod"] = torch.sqrt(1.0 - alphas_cumprod)
        t["sqrt_recip_alphas_cumprod"] = torch.sqrt(1.0 / alphas_cumprod)
        t["sqrt_recipm1_alphas_cumprod"] = torch.sqrt(1.0 / alphas_cumprod - 1)
        post_var = betas * (1.0 - alphas_cumprod_prev) / (1.0 - alphas_cumprod)
        t["posterior_variance"] = post_var
        t["posterior_log_variance_clipped"] = torch.log(post_var.clamp(min=1e-20))
        t["posterior_mean_coef1"] = (
            betas * torch.sqrt(alphas_cumprod_prev) / (1.0 - alphas_cumprod))
        t["posterior_mean_coef2"] = (
            (1.0 - alphas_cumprod_prev) * torch.sqrt(alphas) / (1.0 - alphas_cumprod))

        self._tables_f64 = t
        self._cache = {}  # (device, dtype) -> dict of tables
        for name in self.TABLE_NAMES:
            setattr(self, name, t[name].to(torch.float32))

    def tables(self, device: torch.device, dtype: torch.dtype = torch.float32) -> dict:
        """All tables on `device` as `dtype` (cached)."""
        key = (str(device), dtype)
        if key not in self._cache:
            self._cache[key] = {n: self._tables_f64[n].to(device=device, dtype=dtype)
                                for n in self.TABLE_NAMES}
        return self._cache[key]

    # -- Host-side reverse-process helpers (eager sampler path; the fused
    #    on-device version lives in ops/ (K20) and diffusion/sampler.py). --

    def predict_start_from_noise(self, z_t: torch.Tensor, t, noise: torch.Tensor):
        """x0_hat = sqrt(1/abar_t) z_t - sqrt(1/abar_t - 1) eps  (sampling.py:43-44)."""
        tab = self.tables(z_t.device, z_t.dtype)
        c1 = _gather(tab["sqrt_recip_alphas_cumprod"], t, z_t)
        c2 = _gather(tab["sqrt_recipm1_alphas_cumprod"], t, z_t)
        return c1 * z_t - c2 * noise

    def q_posterior(self, x_start: torch.Tensor, z_t: torch.Tensor, t):
        """mean/var/logvar of q(z_{t-1} | z_t, x0)  (sampling.py:46-53)."""
        tab = self.tables(z_t.device, z_t.dtype)
        mean = (_gather(tab["posterior_mean_coef1"], t, z_t) * x_start
                + _gather(tab["posterior_mean_coef2"], t, z_t) * z_t)
        var = _gather(tab["posterior_variance"], t, z_t)
        logvar = _gather(tab["posterior_log_variance_clipped"], t, z_t)
        return mean, var, logvar


def _gather(table: torch.Tensor, t, like: torch.Tensor) -> torch.Tensor:
    """table[t] broadcast to `like`'s batch shape; t int or (B,) tensor."""
    if isinstance(t, int):
        return table[t]
    t = t.to(table.device)
    out = table.gather(0, t.reshape(-1))
    return out.reshape(t.shape[0], *([1] * (like.ndim - 1)))
