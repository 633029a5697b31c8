"""DDPM ancestral sampler with classifier-free guidance — on-device.

Reference: /root/reference/sampling.py:43-53,116-167 (host-numpy loop, 2
un-jitted forwards per step, blocking cv2 display — defects D5/D6/D10).
Redesign:
  * the whole step (CFG forwards + combine + posterior + reparam sample) runs
    on device; the two CFG forwards are batched into ONE forward over 2B
    (cond_mask = [1]*B + [0]*B),
  * per-step coefficients come from device tables indexed by an on-device
    step counter, so the step is hipGraph-capturable (torch.cuda.CUDAGraph;
    one graph replay per step, zero host work in the loop),
  * supports subsequence sampling (e.g. 256 of 1000 steps) via generalized
    posterior coefficients between consecutive kept timesteps,
  * logsnr fed at step t is logsnr(t/T) — the level the model saw in
    training. `legacy_logsnr=True` reproduces reference D5 (uses
    logsnr((t+1)/T)).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from novel_view_synthesis_3d_amd.diffusion.schedules import (
    DiffusionSchedule, logsnr_schedule_cosine,
)


class DDPMSampler:
    def __init__(self, model, schedule: Optional[DiffusionSchedule] = None,
                 num_steps: int = 1000, guidance_weight: float = 3.0,
                 clip_denoised: bool = True, legacy_logsnr: bool = False,
                 use_graph: bool = False):
        self.model = model
        self.schedule = schedule or DiffusionSchedule(1000)
        self.num_steps = num_steps
        self.w = guidance_weight
        self.clip_denoised = clip_denoised
        self.legacy_logsnr = legacy_logsnr
        self.use_graph = use_graph
        self._graph = None
        self._static = None

    # -----------------------------------------------------------------
    def _step_tables(self, device: torch.device) -> Dict[str, torch.Tensor]:
        """Per-sampling-step coefficient tables (S,), step index 0 = t=T-1.

        For the full sequence these equal the reference's tables
        (sampling.py:28-41); for a subsequence they are the generalized
        DDPM posterior between consecutive kept timesteps.
        """
        sched = self.schedule
        T = sched.timesteps
        S = self.num_steps
        # kept timesteps, descending; S=T -> [T-1, ..., 0]
        ts = torch.linspace(T - 1, 0, S).round().long()
        abar = sched._tables_f64["alphas_cumprod"]
        abar_t = abar[ts]
        abar_prev = torch.cat([abar[ts[1:]], torch.ones(1, dtype=torch.float64)])
        alpha_eff = abar_t / abar_prev
        beta_eff = 1.0 - alpha_eff
        post_var = beta_eff * (1.0 - abar_prev) / (1.0 - abar_t)
        tab = {
            "sqrt_recip_abar": torch.sqrt(1.0 / abar_t),
            "sqrt_recipm1_abar": torch.sqrt(1.0 / abar_t - 1.0),
            "mean_c1": beta_eff * torch.sqrt(abar_prev) / (1.0 - abar_t),
            "mean_c2": (1.0 - abar_prev) * torch.sqrt(alpha_eff) / (1.0 - abar_t),
            "sigma": torch.exp(0.5 * torch.log(post_var.clamp(min=1e-20))),
        }
        # noise is zeroed at the final step (t==0) — reference sampling.py:147
        # (fixing D6: the mask is per-step, not len(z)-shaped)
        tab["sigma"] = tab["sigma"] * (ts > 0).to(torch.float64)
        u = (ts + 1 if self.legacy_logsnr else ts).to(torch.float64) / T
        tab["logsnr"] = torch.tensor(
            [logsnr_schedule_cosine(float(x)) for x in u], dtype=torch.float64)
        return {k: v.to(device=device, dtype=torch.float32) for k, v in tab.items()}

    # -----------------------------------------------------------------
    def _make_cond2(self, cond: Dict[str, torch.Tensor]):
        """Duplicate conditioning along batch for the batched CFG forward."""
        c2 = {k: torch.cat([v, v], dim=0) for k, v in cond.items()
              if k in ("x", "R1", "t1", "R2", "t2", "K")}
        B = cond["x"].shape[0]
        mask = torch.cat([torch.ones(B, device=cond["x"].device),
                          torch.zeros(B, device=cond["x"].device)])
        return c2, mask

    def _step(self, z: torch.Tensor, idx: torch.Tensor, cond2, mask, tab,
              pose_cache=None) -> None:
        """One in-place ancestral step; everything device-side."""
        B = z.shape[0]
        logsnr = tab["logsnr"].index_select(0, idx).expand(2 * B)
        batch = dict(cond2)
        batch["z"] = torch.cat([z, z], dim=0)
        batch["logsnr"] = logsnr
        if z.is_cuda:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = self.model(batch, mask, pose_cache=pose_cache)
        else:
            out = self.model(batch, mask, pose_cache=pose_cache)
        out = out.to(torch.float32)
        eps = (1.0 + self.w) * out[:B] - self.w * out[B:]

        c = {k: tab[k].index_select(0, idx) for k in
             ("sqrt_recip_abar", "sqrt_recipm1_abar", "mean_c1", "mean_c2",
              "sigma")}
        x0 = c["sqrt_recip_abar"] * z - c["sqrt_recipm1_abar"] * eps
        if self.clip_denoised:
            x0 = x0.clamp(-1.0, 1.0)
        mean = c["mean_c1"] * x0 + c["mean_c2"] * z
        noise = torch.randn_like(z)
        z.copy_(mean + c["sigma"] * noise)
        idx.add_(1)

    # -----------------------------------------------------------------
    @torch.no_grad()
    def sample(self, cond: Dict[str, torch.Tensor],
               z_init: Optional[torch.Tensor] = None,
               generator: Optional[torch.Generator] = None) -> torch.Tensor:
        """Generate target views for `cond` (x, R1, t1, R2, t2, K on device).

        Returns (B, H, W, 3) in [-1, 1] (clipped x0 convention of the
        reference: the final step's mean with t=0 noise masked).
        """
        was_training = self.model.training
        self.model.eval()
        try:
            x = cond["x"]
            device = x.device
            z = (z_init.clone() if z_init is not None
                 else torch.empty_like(x).normal_(generator=generator))
            cond2, mask = self._make_cond2(cond)
            tab = self._step_tables(device)
            idx = torch.zeros(1, dtype=torch.long, device=device)

            # Pose conditioning is step-invariant: compute once, outside any
            # graph capture (fixes both the 2x redundant work per step and
            # capture-unsupported ops in the camera math).
            cp = self.model.ConditioningProcessor_0
            if device.type == "cuda":
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    pose_cache = cp.pose_features(cond2, mask)
            else:
                pose_cache = cp.pose_features(cond2, mask)

            if self.use_graph and device.type == "cuda":
                self._run_graphed(z, idx, cond2, mask, tab, pose_cache)
            else:
                for _ in range(self.num_steps):
                    self._step(z, idx, cond2, mask, tab, pose_cache)
            return z
        finally:
            self.model.train(was_training)

    # -----------------------------------------------------------------
    def _run_graphed(self, z, idx, cond2, mask, tab, pose_cache) -> None:
        """Capture one sampler step as a hipGraph and replay it num_steps
        times. Per-step state (z, idx) lives in the captured buffers; RNG is
        graph-safe (torch captures the philox offset)."""
        # warm up at most num_steps steps: indexing past the step tables is a
        # device-side assert, and with nothing left to replay the graph is
        # pointless anyway
        n_warm = min(2, self.num_steps)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):  # warmup, required before capture
            for _ in range(n_warm):
                self._step(z, idx, cond2, mask, tab, pose_cache)
        torch.cuda.current_stream().wait_stream(s)
        done_warmup = int(idx.item())
        if self.num_steps - done_warmup <= 0:
            return

        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            # capture RECORDS the step without executing it; z/idx advance
            # only on replay
            self._step(z, idx, cond2, mask, tab, pose_cache)
        for _ in range(self.num_steps - done_warmup):
            graph.replay()
