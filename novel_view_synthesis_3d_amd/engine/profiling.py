"""Profiling subsystem (the reference has none — SURVEY.md §5.1).

Two layers:
  * `torch_profile(...)` — torch.profiler (kineto/roctracer) trace of a few
    training steps, chrome-trace exported for chrome://tracing / perfetto.
  * rocprofv3 is driven externally (per-kernel stats + PMC counters):
      cd /tmp && rocprofv3 --kernel-trace --stats -d out -- python bench.py ...
    Committed summaries live in profiles/.
"""

from __future__ import annotations

import contextlib
import os
from typing import Optional

import torch


@contextlib.contextmanager
def torch_profile(out_dir: str = "./profiles", enabled: bool = True,
                  with_stack: bool = False):
    """Context manager: profile the enclosed steps, write a chrome trace."""
    if not enabled:
        yield None
        return
    os.makedirs(out_dir, exist_ok=True)
    acts = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=acts, with_stack=with_stack,
                                record_shapes=False) as prof:
        yield prof
    path = os.path.join(out_dir, "torch_trace.json")
    prof.export_chrome_trace(path)
    print(f"[profiling] wrote {path}")
    print(prof.key_averages().table(sort_by="self_cuda_time_total",
                                    row_limit=25))


def profile_training(trainer, steps: int = 5, out_dir: str = "./profiles"):
    """Profile `steps` training steps of a Trainer."""
    for _ in range(3):  # warmup outside the profile
        trainer.train_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    with torch_profile(out_dir):
        for _ in range(steps):
            trainer.train_step()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
