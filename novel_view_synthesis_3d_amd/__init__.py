"""novel_view_synthesis_3d_amd — an MI355X-native 3DiM novel-view-synthesis engine.

A from-scratch re-design of the capabilities of
`shiveshkhaitan/novel_view_synthesis_3d` (JAX/FLAX, reference mounted at
/root/reference) for AMD Instinct MI355X (gfx950):

* pose-conditional X-UNet denoiser (`models.xunet.XUNet`) operating on a
  2-frame stack (clean source view + noisy target view), NHWC-with-frame-axis
  activations chosen for CDNA4 memory coalescing,
* hand-written HIP/CDNA4 kernels for the hot ops (`ops/hip/`), with eager
  PyTorch reference implementations (`ops/reference.py`) kept as the
  numerics oracle and CPU path,
* DDPM diffusion schedules + forward noising + CFG ancestral sampler
  (`diffusion/`),
* true data-parallel training over RCCL/xGMI (`parallel/ddp.py`) — the
  reference's jax.pmap ensemble quirk replaced by rank-0
  broadcast init + bucketed gradient all-reduce,
* SRN/ShapeNet-style scene dataset + synthetic generator (`data/`),
* Trainer / checkpointing with a FLAX-compatible parameter layout map
  (`engine/`).
"""

__version__ = "0.1.0"

from novel_view_synthesis_3d_amd.config import XUNetConfig, TrainConfig  # noqa: F401
