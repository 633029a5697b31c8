"""Weight initializers matching FLAX semantics.

The reference model's params come from flax defaults: Conv/Dense use
lecun_normal (variance_scaling(1.0, 'fan_in', 'truncated_normal')), bias
zeros; output convs use variance_scaling(0., ...) i.e. zeros
(/root/reference/model/xunet.py:11-12). jax's truncated_normal draws from a
stddev-1 normal truncated to [-2, 2] and rescales by 1/0.87962566 so the
*post-truncation* stddev is 1; variance_scaling then multiplies by
sqrt(scale/fan_in).
"""

from __future__ import annotations

import math

import torch

# stddev of a unit normal truncated to [-2, 2]
_TRUNC_STD = 0.8796256610342398


def variance_scaling_(tensor: torch.Tensor, fan_in: int, scale: float = 1.0,
                      generator=None) -> torch.Tensor:
    """In-place variance-scaling truncated-normal init (flax-compatible)."""
    if scale == 0.0:
        with torch.no_grad():
            return tensor.zero_()
    std = math.sqrt(scale / fan_in)
    with torch.no_grad():
        # true truncated normal (inverse-CDF, like jax truncated_normal) —
        # clamping instead would pile mass at +-2 and overshoot the std ~9%
        torch.nn.init.trunc_normal_(tensor, mean=0.0, std=1.0,
                                    a=-2.0, b=2.0, generator=generator)
        tensor.mul_(std / _TRUNC_STD)
    return tensor


def lecun_normal_(tensor: torch.Tensor, fan_in: int, generator=None) -> torch.Tensor:
    return variance_scaling_(tensor, fan_in, 1.0, generator=generator)


def normal_(tensor: torch.Tensor, std: float, generator=None) -> torch.Tensor:
    with torch.no_grad():
        tensor.normal_(0.0, std, generator=generator)
    return tensor
