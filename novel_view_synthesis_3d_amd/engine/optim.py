"""FusedAdam — single-launch multi-tensor Adam on MI355X (K18).

State layout matches torch.optim.Adam ('step', 'exp_avg', 'exp_avg_sq'),
so checkpoints interchange with the eager optimizer. Falls back to
torch.optim.Adam math on CPU / when the extension is unavailable.
"""

from __future__ import annotations

import math
from typing import Optional

import torch


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-4, betas=(0.9, 0.999), eps=1e-8):
        defaults = dict(lr=lr, betas=betas, eps=eps)
        super().__init__(params, defaults)
        self._plans = {}  # param-group index -> AdamPlan

    def _hip(self):
        try:
            from novel_view_synthesis_3d_amd.ops import hip_ops
            return hip_ops
        except Exception as e:
            import os
            if os.environ.get("NVS3D_ALLOW_EAGER_GPU", "0") == "1":
                return None
            # same fail-loudly policy as ops dispatch: a GPU box must not
            # silently fall back to the slow eager Adam
            raise RuntimeError(
                f"FusedAdam on GPU but nvs3d_hip extension unavailable: {e}"
            ) from e

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            # lazy state init (torch.optim.Adam-compatible layout)
            for p in params:
                st = self.state[p]
                if len(st) == 0:
                    st["step"] = torch.tensor(0.0)
                    st["exp_avg"] = torch.zeros_like(p)
                    st["exp_avg_sq"] = torch.zeros_like(p)
            for p in params:
                self.state[p]["step"] += 1
            steps = {int(self.state[p]["step"].item()) for p in params}
            # the fused plan applies ONE bias correction to the whole group;
            # a param that skipped steps (no grad some iterations) would need
            # per-param correction — fail loudly instead of silently drifting
            assert len(steps) == 1, (
                f"FusedAdam group {gi} has divergent per-param step counts "
                f"{sorted(steps)}; per-group fused bias correction is invalid")
            step = steps.pop()
            b1, b2 = group["betas"]

            hip = self._hip() if params[0].is_cuda else None
            if hip is not None:
                tuples = [(p, p.grad, self.state[p]["exp_avg"],
                           self.state[p]["exp_avg_sq"]) for p in params]
                plan = self._plans.get(gi)
                if plan is None or not plan.matches(tuples):
                    plan = hip.AdamPlan(tuples, params[0].device)
                    self._plans[gi] = plan
                hip.fused_adam_step(plan, group["lr"], b1, b2,
                                    group["eps"], step)
            else:  # eager fallback (CPU path / parity tests)
                bc1 = 1 - b1 ** step
                bc2 = 1 - b2 ** step
                for p in params:
                    st = self.state[p]
                    g = p.grad
                    st["exp_avg"].mul_(b1).add_(g, alpha=1 - b1)
                    st["exp_avg_sq"].mul_(b2).addcmul_(g, g, value=1 - b2)
                    denom = (st["exp_avg_sq"] / bc2).sqrt_().add_(group["eps"])
                    p.addcdiv_(st["exp_avg"] / bc1, denom, value=-group["lr"])
        return loss
