"""Data-parallel training over RCCL/xGMI.

The reference's jax.pmap "distributed" training is an *ensemble*, not DP
(per-device independent init, replicated batch, no
gradient reduction). This module implements what the reference README claims:
true data parallelism — one process per GPU (torchrun), rank-0 parameter
broadcast at init, per-rank batch shards, and bucketed gradient all-reduce
overlapped with backward.

MI355X sizing: xGMI is 7 point-to-point links x ~153 GB/s
per GPU; RCCL rings are per-link bound, so we use a few large buckets
(default 40 MB — the full-config X-UNet's biggest grads are ~37 MB conv
weights) launched as soon as their grads accumulate, so communication rides
under the remaining backward compute.

Backend: "nccl" (RCCL on ROCm) on GPU; "gloo" on CPU (multi-process CPU
tests run the identical code path).
"""

from __future__ import annotations

import os
from contextlib import contextmanager
from typing import List, Optional

import torch
import torch.distributed as dist


def distributed_info():
    """(rank, world_size, local_rank) from env; (0, 1, 0) if not launched
    via torchrun."""
    return (int(os.environ.get("RANK", "0")),
            int(os.environ.get("WORLD_SIZE", "1")),
            int(os.environ.get("LOCAL_RANK", "0")))


def init_distributed(device_type: Optional[str] = None):
    """Initialize the default process group if torchrun env vars are present.

    Returns (rank, world_size, local_rank). Backend: nccl (=RCCL) when CUDA
    is available, else gloo.
    """
    rank, world, local_rank = distributed_info()
    if world > 1 and not dist.is_initialized():
        if device_type is None:
            device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = "nccl" if device_type == "cuda" else "gloo"
        if device_type == "cuda":
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)
    return rank, world, local_rank


class _Bucket:
    __slots__ = ("params", "numel", "buffer", "ready", "handle", "launched")

    def __init__(self):
        self.params: List[torch.nn.Parameter] = []
        self.numel = 0
        self.buffer: Optional[torch.Tensor] = None
        self.ready = 0
        self.handle = None
        self.launched = False


class DataParallelEngine:
    """Bucketed gradient all-reduce with PERSISTENT FLAT GRAD STORAGE:
    every param's .grad is a view into its bucket's contiguous fp32 buffer
    (what torch DDP calls gradient_as_bucket_view), so the all-reduce runs
    in place on the gradients themselves — zero staging copies per step.
    With RCCL the reduction uses ReduceOp.AVG (no separate 1/N pass).

    Usage:
        engine = DataParallelEngine(model, bucket_mb=40)
        for step ...:
            engine.zero_flags()      # zeroes the flat buffers (= the grads)
            loss.backward()          # hooks launch async in-place reduces
            engine.finish()          # wait (+ 1/N on gloo)
            optimizer.step()
    """

    def __init__(self, model: torch.nn.Module, bucket_mb: float = 40.0,
                 process_group=None):
        self.model = model
        self.group = process_group
        self.world = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world > 1
        self._sync = True
        self._hooks = []
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        if not self.enabled:
            return
        # ReduceOp.AVG is an RCCL-native fused sum+scale; gloo needs SUM+mul
        self._avg = dist.get_backend(self.group) == "nccl"

        # Identical start state on every rank (reference defect D3 fix).
        with torch.no_grad():
            for p in model.parameters():
                dist.broadcast(p.data, src=0, group=self.group)

        # Buckets in REVERSE parameter order (grads become ready roughly in
        # reverse registration order during backward).
        params = [p for p in model.parameters() if p.requires_grad]
        cap = int(bucket_mb * 1e6 / 4)  # fp32 elements
        bucket = _Bucket()
        for p in reversed(params):
            if bucket.numel > 0 and bucket.numel + p.numel() > cap:
                self.buckets.append(bucket)
                bucket = _Bucket()
            bucket.params.append(p)
            bucket.numel += p.numel()
            self._param_bucket[p] = bucket
        if bucket.numel:
            self.buckets.append(bucket)

        # flat buffers; each grad becomes a view (autograd accumulates into
        # existing .grad in place, so the views persist across steps)
        for b in self.buckets:
            b.buffer = torch.zeros(b.numel, dtype=torch.float32,
                                   device=b.params[0].device)
            off = 0
            for p in b.params:
                n = p.numel()
                p.grad = b.buffer[off:off + n].view(p.shape)
                off += n
        self._buffers = [b.buffer for b in self.buckets]

        for p in params:
            h = p.register_post_accumulate_grad_hook(self._on_grad)
            self._hooks.append(h)

    # -----------------------------------------------------------------
    def _on_grad(self, p: torch.nn.Parameter) -> None:
        if not (self.enabled and self._sync):
            return
        b = self._param_bucket[p]
        b.ready += 1
        if b.ready == len(b.params):
            self._launch(b)

    def _launch(self, b: _Bucket) -> None:
        op = dist.ReduceOp.AVG if self._avg else dist.ReduceOp.SUM
        b.handle = dist.all_reduce(b.buffer, op=op, group=self.group,
                                   async_op=True)
        b.launched = True

    def zero_flags(self) -> None:
        """Reset per-step state AND zero the flat grad buffers (replaces
        optimizer.zero_grad — .grad views must never be detached)."""
        for b in self.buckets:
            b.ready = 0
            b.handle = None
            b.launched = False
        if self.enabled:
            torch._foreach_zero_(self._buffers)

    def finish(self) -> None:
        """Wait for all in-place reductions (launch any bucket whose params
        produced no grad hook this step — its buffer holds zeros/partials)."""
        if not (self.enabled and self._sync):
            return
        for b in self.buckets:
            if not b.launched:
                self._launch(b)
        inv = 1.0 / self.world
        for b in self.buckets:
            b.handle.wait()
            if not self._avg:
                b.buffer.mul_(inv)

    def reduce_all(self) -> None:
        """Synchronous whole-model reduce (used after a hipGraph replay,
        where the in-graph backward cannot launch RCCL ops)."""
        if not self.enabled:
            return
        for b in self.buckets:
            b.ready = len(b.params)
            b.launched = False
        for b in self.buckets:
            self._launch(b)
        inv = 1.0 / self.world
        for b in self.buckets:
            b.handle.wait()
            if not self._avg:
                b.buffer.mul_(inv)

    @contextmanager
    def no_sync(self):
        self._sync = False
        try:
            yield
        finally:
            self._sync = True

    def all_reduce_scalar(self, value: torch.Tensor, avg: bool = True):
        """Cross-rank scalar reduce for logging."""
        if not self.enabled:
            return value
        v = value.detach().clone()
        dist.all_reduce(v, op=dist.ReduceOp.SUM, group=self.group)
        return v / self.world if avg else v

    def max_scalar(self, value: float) -> float:
        if not self.enabled:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        if torch.cuda.is_available():
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=self.group)
        return float(t.item())
