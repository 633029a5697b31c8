"""X-UNet building blocks.

Each module mirrors one FLAX module of the reference
(/root/reference/model/xunet.py) in math and in *construction order*, so the
parameter tree maps 1:1 onto the FLAX checkpoint layout (see
engine/checkpoint.py). Activations are (B, F=2, H, W, C) contiguous.

Compute goes through novel_view_synthesis_3d_amd.ops — hand-written CDNA4 HIP
kernels on MI355X, eager torch reference elsewhere.
"""

from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from novel_view_synthesis_3d_amd import ops
from novel_view_synthesis_3d_amd.utils.init import lecun_normal_


# ---------------------------------------------------------------------------
# Leaf modules
# ---------------------------------------------------------------------------

class FrameConv(nn.Module):
    """Per-frame 3x3 'SAME' conv — flax nn.Conv(kernel=(1,3,3), strides=(1,s,s))
    (/root/reference/model/xunet.py:81,85,199-202,229,276).

    Weight stored (Cout, 3, 3, Cin) contiguous (OHWI = channels-last layout
    of an OIHW conv weight, and the layout the CDNA4 implicit-GEMM conv
    kernel consumes); flax kernel layout is (1, 3, 3, Cin, Cout).
    """

    flax_type = "Conv"

    def __init__(self, cin: int, cout: int, stride: int = 1,
                 zero_init: bool = False):
        super().__init__()
        self.cin, self.cout, self.stride = cin, cout, stride
        self.weight = nn.Parameter(torch.empty(cout, 3, 3, cin))
        self.bias = nn.Parameter(torch.zeros(cout))
        if zero_init:
            nn.init.zeros_(self.weight)
        else:
            lecun_normal_(self.weight, fan_in=cin * 9)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.frame_conv3x3(x, self.weight, self.bias, self.stride)

    def flax_leaves(self):
        return [
            ("kernel", self.weight,
             lambda w: w.permute(1, 2, 3, 0).unsqueeze(0),      # -> (1,3,3,Cin,Cout)
             lambda f: torch.as_tensor(f).squeeze(0).permute(3, 0, 1, 2)),
            ("bias", self.bias, lambda b: b, lambda f: torch.as_tensor(f)),
        ]


class Dense(nn.Module):
    """flax nn.Dense: y = x @ kernel + bias; kernel (in, out)."""

    flax_type = "Dense"

    def __init__(self, cin: int, cout: int, zero_init: bool = False):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(cout, cin))
        self.bias = nn.Parameter(torch.zeros(cout))
        if zero_init:
            nn.init.zeros_(self.weight)
        else:
            lecun_normal_(self.weight, fan_in=cin)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.linear(x, self.weight, self.bias)

    def flax_leaves(self):
        return [
            ("kernel", self.weight, lambda w: w.t(),
             lambda f: torch.as_tensor(f).t()),
            ("bias", self.bias, lambda b: b, lambda f: torch.as_tensor(f)),
        ]


class DenseGeneral(nn.Module):
    """flax nn.DenseGeneral(features=(heads, head_dim)): kernel (C, h, d).

    Used for the q/k/v projections (/root/reference/model/xunet.py:100-102).
    Implemented as a single GEMM C -> h*d + reshape.
    """

    flax_type = "DenseGeneral"

    def __init__(self, cin: int, heads: int, head_dim: int):
        super().__init__()
        self.heads, self.head_dim = heads, head_dim
        self.weight = nn.Parameter(torch.empty(heads * head_dim, cin))
        self.bias = nn.Parameter(torch.zeros(heads * head_dim))
        lecun_normal_(self.weight, fan_in=cin)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight, self.bias)
        return y.reshape(*y.shape[:-1], self.heads, self.head_dim)

    def flax_leaves(self):
        h, d = self.heads, self.head_dim
        return [
            ("kernel", self.weight,
             lambda w: w.t().reshape(w.shape[1], h, d),
             lambda f: torch.as_tensor(f).reshape(-1, h * d).t()),
            ("bias", self.bias, lambda b: b.reshape(h, d),
             lambda f: torch.as_tensor(f).reshape(h * d)),
        ]


class JointGroupNorm(nn.Module):
    """GroupNorm with stats jointly over both frames + space
    (/root/reference/model/xunet.py:46-52). flax nesting: the reference wraps
    flax nn.GroupNorm in its own GroupNorm module, so the flax path is
    'GroupNorm_i/GroupNorm_0/{scale,bias}'.

    groups = min(32, C) fixes reference defect D7 (ch < 32 tiny configs).
    """

    flax_type = "GroupNorm"
    EPS = 1e-6  # flax nn.GroupNorm default

    def __init__(self, channels: int):
        super().__init__()
        self.channels = channels
        g = min(32, channels)
        while channels % g:  # largest divisor of C that is <= 32
            g -= 1
        self.groups = g
        self.scale = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x: torch.Tensor, film=None, silu: bool = False,
                p_drop: float = 0.0) -> torch.Tensor:
        return ops.joint_groupnorm(x, self.scale, self.bias, self.groups,
                                   self.EPS, film, silu, p_drop)

    def flax_leaves(self):
        return [
            ("GroupNorm_0/scale", self.scale, lambda s: s,
             lambda f: torch.as_tensor(f)),
            ("GroupNorm_0/bias", self.bias, lambda b: b,
             lambda f: torch.as_tensor(f)),
        ]


class FiLM(nn.Module):
    """Feature-wise linear modulation (/root/reference/model/xunet.py:54-61):
    Dense(2*features)(silu(emb)) -> packed (.., 2*features) scale|shift. The
    modulation itself is fused into the GroupNorm kernel
    (ops.joint_groupnorm), which consumes the packed tensor strided — no
    split/copy. The caller passes emb ALREADY silu'd (XUNet caches silu(emb)
    once per level instead of once per block)."""

    flax_type = "FiLM"

    def __init__(self, emb_ch: int, features: int):
        super().__init__()
        self.features = features
        self.Dense_0 = Dense(emb_ch, 2 * features)

    def packed(self, emb_silu: torch.Tensor) -> torch.Tensor:
        return self.Dense_0(emb_silu)


# ---------------------------------------------------------------------------
# Blocks
# ---------------------------------------------------------------------------

class ResnetBlock(nn.Module):
    """BigGAN-style residual block over frames
    (/root/reference/model/xunet.py:63-92):

        h = silu(GN(h_in)); [resample h and h_in]; h = conv1(h)
        h = silu(FiLM(GN(h), emb)); dropout; h = conv2_zeroinit(h)
        h_in = dense(h_in) if C changed
        return (h + h_in)/sqrt(2)

    FLAX creation order: GroupNorm_0, Conv_0, FiLM_0, GroupNorm_1, Conv_1,
    [Dense_0] (FiLM is constructed before its GroupNorm argument evaluates).
    """

    flax_type = "ResnetBlock"

    def __init__(self, cin: int, emb_ch: int, features: Optional[int] = None,
                 dropout: float = 0.0, resample: Optional[str] = None):
        super().__init__()
        features = cin if features is None else features
        self.cin, self.features = cin, features
        self.dropout_rate = dropout
        self.resample = resample
        self.GroupNorm_0 = JointGroupNorm(cin)
        self.Conv_0 = FrameConv(cin, features)
        self.FiLM_0 = FiLM(emb_ch, features)
        self.GroupNorm_1 = JointGroupNorm(features)
        self.Conv_1 = FrameConv(features, features, zero_init=True)
        if cin != features:
            self.Dense_0 = Dense(cin, features)
        else:
            self.Dense_0 = None

    def forward(self, h_in: torch.Tensor, emb_silu: torch.Tensor
                ) -> torch.Tensor:
        h = self.GroupNorm_0(h_in, silu=True)
        if self.resample == "up":
            h = ops.nearest_upsample2x(h)
            h_in = ops.nearest_upsample2x(h_in)
        elif self.resample == "down":
            h = ops.avgpool_downsample2x(h)
            h_in = ops.avgpool_downsample2x(h_in)
        h = self.Conv_0(h)
        film = self.FiLM_0.packed(emb_silu)
        # the whole inter-conv segment is ONE kernel: GN+FiLM+SiLU+dropout
        h = self.GroupNorm_1(h, film=film, silu=True,
                             p_drop=(self.dropout_rate if self.training
                                     else 0.0))
        if self.Dense_0 is not None:
            h_in = self.Dense_0(h_in)
        # residual + 1/sqrt(2) fused into the conv epilogue (xunet.py:92)
        return ops.frame_conv3x3_residual(
            h, self.Conv_1.weight, self.Conv_1.bias,
            h_in.to(h.dtype) if h_in.dtype != h.dtype else h_in,
            ops.SQRT_HALF)


class AttnLayer(nn.Module):
    """q/k/v projections + multi-head attention
    (/root/reference/model/xunet.py:94-103). NOTE: no output projection —
    the reference comments it out (xunet.py:126).

    The three DenseGenerals are fused into ONE projection GEMM at call time
    (weights concatenated; q/k/v become strided views of the (B,L,3C) output
    — the MFMA attention kernel consumes the views directly)."""

    flax_type = "AttnLayer"

    def __init__(self, channels: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = channels // heads
        head_dim = self.head_dim
        self.DenseGeneral_0 = DenseGeneral(channels, heads, head_dim)  # q
        self.DenseGeneral_1 = DenseGeneral(channels, heads, head_dim)  # k
        self.DenseGeneral_2 = DenseGeneral(channels, heads, head_dim)  # v

    def project_qkv(self, x: torch.Tensor):
        """(B,L,C) -> q,k,v (B,L,h,d) strided views of one fused GEMM."""
        C = x.shape[-1]
        w = torch.cat([self.DenseGeneral_0.weight,
                       self.DenseGeneral_1.weight,
                       self.DenseGeneral_2.weight], dim=0)
        b = torch.cat([self.DenseGeneral_0.bias, self.DenseGeneral_1.bias,
                       self.DenseGeneral_2.bias], dim=0)
        y = ops.linear(x, w, b)  # (B, L, 3C)
        hd = (self.heads, self.head_dim)
        return (y[..., :C].unflatten(-1, hd),
                y[..., C:2 * C].unflatten(-1, hd),
                y[..., 2 * C:].unflatten(-1, hd))

    def forward(self, q: torch.Tensor, kv: torch.Tensor) -> torch.Tensor:
        B, L, C = q.shape
        if q is kv:
            qq, kk, vv = self.project_qkv(q)
        else:
            qq = self.DenseGeneral_0(q)
            kk = self.DenseGeneral_1(kv)
            vv = self.DenseGeneral_2(kv)
        out = ops.attention(qq, kk, vv)
        return out.reshape(B, L, C)


class AttnBlock(nn.Module):
    """Self- or cross-frame attention block
    (/root/reference/model/xunet.py:105-127). ONE AttnLayer is shared by both
    frames' calls (reference creates a single attn_layer and applies it twice).
    """

    flax_type = "AttnBlock"

    def __init__(self, channels: int, heads: int, attn_type: str):
        super().__init__()
        assert attn_type in ("self", "cross")
        self.attn_type = attn_type
        self.GroupNorm_0 = JointGroupNorm(channels)
        self.AttnLayer_0 = AttnLayer(channels, heads)

    def forward(self, h_in: torch.Tensor) -> torch.Tensor:
        B, Fr, H, W, C = h_in.shape
        h = self.GroupNorm_0(h_in)
        # BOTH frames batched: ONE fused QKV projection (the reference
        # projects q and k/v separately per frame per call — 6 GEMMs where 1
        # suffices) and ONE attention launch; cross-frame attention is the
        # same launch with the kernel pairing batch b with k/v of b^1.
        h2 = h.reshape(B * Fr, H * W, C)
        q, k, v = self.AttnLayer_0.project_qkv(h2)
        o = ops.attention(q, k, v, kv_swap=(self.attn_type == "cross"))
        h = o.reshape(B, Fr, H, W, C)
        return ops.residual_scale_add(h, h_in)


class XUNetBlock(nn.Module):
    """ResnetBlock optionally followed by self+cross attention
    (/root/reference/model/xunet.py:129-140)."""

    flax_type = "XUNetBlock"

    def __init__(self, cin: int, emb_ch: int, features: int, heads: int,
                 dropout: float, use_attn: bool):
        super().__init__()
        self.ResnetBlock_0 = ResnetBlock(cin, emb_ch, features=features,
                                         dropout=dropout)
        if use_attn:
            self.AttnBlock_0 = AttnBlock(features, heads, "self")
            self.AttnBlock_1 = AttnBlock(features, heads, "cross")
        else:
            self.AttnBlock_0 = None
            self.AttnBlock_1 = None

    def forward(self, x: torch.Tensor, emb: torch.Tensor) -> torch.Tensor:
        h = self.ResnetBlock_0(x, emb)
        if self.AttnBlock_0 is not None:
            h = self.AttnBlock_0(h)
            h = self.AttnBlock_1(h)
        return h
