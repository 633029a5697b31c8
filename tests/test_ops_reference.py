"""Eager reference-op tests (these ops are the oracle for the HIP kernels)."""

import math

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from novel_view_synthesis_3d_amd.ops import reference as ref


def test_joint_groupnorm_stats_span_frames():
    """Stats must span BOTH frames (reference xunet.py:46-52): normalizing a
    tensor whose two frames have different scales must NOT whiten each frame
    separately."""
    torch.manual_seed(0)
    B, Fr, H, W, C = 2, 2, 4, 4, 8
    x = torch.randn(B, Fr, H, W, C)
    x[:, 1] *= 10.0  # frame 1 is 10x hotter
    g = torch.ones(C)
    b = torch.zeros(C)
    out = ref.joint_groupnorm(x, g, b, groups=4)
    # manual: per (batch, group) over (F,H,W,C/g)
    xg = x.reshape(B, Fr, H, W, 4, 2)
    mean = xg.mean(dim=(1, 2, 3, 5), keepdim=True)
    var = xg.var(dim=(1, 2, 3, 5), unbiased=False, keepdim=True)
    manual = ((xg - mean) / torch.sqrt(var + 1e-6)).reshape(B, Fr, H, W, C)
    assert torch.allclose(out, manual, atol=1e-5)
    # frame 1 must remain hotter after joint normalization
    assert out[:, 1].std() > 3 * out[:, 0].std()


def test_joint_groupnorm_film_silu_fusion():
    torch.manual_seed(1)
    x = torch.randn(2, 2, 4, 4, 8)
    gamma = torch.randn(8) * 0.1 + 1
    beta = torch.randn(8) * 0.1
    scale = torch.randn(2, 2, 4, 4, 8) * 0.1
    shift = torch.randn(2, 2, 4, 4, 8) * 0.1
    fused = ref.joint_groupnorm(x, gamma, beta, 4,
                                film=torch.cat([scale, shift], dim=-1),
                                silu=True)
    base = ref.joint_groupnorm(x, gamma, beta, 4)
    manual = F.silu(base * (1 + scale) + shift)
    assert torch.allclose(fused, manual, atol=1e-5)


def test_frame_conv_same_padding_stride1():
    torch.manual_seed(2)
    x = torch.randn(1, 2, 8, 8, 4)
    w = torch.randn(6, 3, 3, 4) * 0.1  # OHWI layout
    b = torch.zeros(6)
    y = ref.frame_conv3x3(x, w, b)
    assert y.shape == (1, 2, 8, 8, 6)
    # compare against direct conv2d on each frame
    for f in range(2):
        xf = x[0, f].permute(2, 0, 1)[None]
        yf = F.conv2d(xf, w.permute(0, 3, 1, 2), b,
                      padding=1)[0].permute(1, 2, 0)
        assert torch.allclose(y[0, f], yf, atol=1e-5)


@pytest.mark.parametrize("stride,H", [(2, 8), (4, 16), (8, 16), (1, 8)])
def test_frame_conv_same_padding_strided(stride, H):
    """FLAX SAME semantics: out = ceil(H/s), asymmetric padding."""
    torch.manual_seed(3)
    x = torch.randn(1, 2, H, H, 3)
    w = torch.randn(5, 3, 3, 3) * 0.1  # OHWI
    y = ref.frame_conv3x3(x, w, None, stride=stride)
    assert y.shape[2] == -(-H // stride)
    assert y.shape == (1, 2, -(-H // stride), -(-H // stride), 5)


def test_attention_matches_sdpa():
    torch.manual_seed(4)
    B, L, h, d = 2, 16, 4, 8
    q, k, v = (torch.randn(B, L, h, d) for _ in range(3))
    out = ref.attention(q, k, v)
    expect = F.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3), v.permute(0, 2, 1, 3)
    ).permute(0, 2, 1, 3)
    assert torch.allclose(out, expect, atol=1e-5)


def test_up_down_sampling():
    x = torch.arange(2 * 2 * 2 * 2 * 1, dtype=torch.float32).reshape(1, 2, 2, 2, 2)
    up = ref.nearest_upsample2x(x)
    assert up.shape == (1, 2, 4, 4, 2)
    assert torch.allclose(up[0, 0, 0, 0], up[0, 0, 1, 1])  # 2x2 blocks equal
    down = ref.avgpool_downsample2x(up)
    assert torch.allclose(down, x, atol=1e-6)


def test_residual_scale_add():
    a, b = torch.randn(3, 2, 4, 4, 5), torch.randn(3, 2, 4, 4, 5)
    assert torch.allclose(ref.residual_scale_add(a, b),
                          (a + b) / math.sqrt(2), atol=1e-7)


def test_posenc_ddpm():
    t = torch.tensor([0.0, 0.5, 1.0])
    e = ref.posenc_ddpm(t, emb_ch=32, max_time=1.0)
    assert e.shape == (3, 32)
    # t=0 -> sin part 0, cos part 1
    assert torch.allclose(e[0, :16], torch.zeros(16), atol=1e-7)
    assert torch.allclose(e[0, 16:], torch.ones(16), atol=1e-7)
    # frequency 0 is 1000*t (reference multiplies by 1000/max_time)
    assert e[1, 0] == pytest.approx(math.sin(500.0), abs=1e-4)


def test_squash_logsnr():
    l = torch.tensor([-25.0, -20.0, 0.0, 20.0, 25.0])
    s = ref.squash_logsnr(l)
    assert s[0] == pytest.approx(s[1])  # clipped
    assert s[2] == pytest.approx(0.5, abs=1e-6)  # atan(1)*2/pi
    assert 0.0 < s[4] < 1e-4


def test_posenc_nerf_dims_and_values():
    x = torch.randn(2, 4, 4, 3)
    p15 = ref.posenc_nerf(x, 0, 15)
    p8 = ref.posenc_nerf(x, 0, 8)
    assert p15.shape[-1] == 3 * (1 + 2 * 15)  # 93
    assert p8.shape[-1] == 3 * (1 + 2 * 8)    # 51
    # identity part first
    assert torch.allclose(p15[..., :3], x)
    # sin(x*2^0) next
    assert torch.allclose(p15[..., 3:6], torch.sin(x), atol=1e-6)
    # phase-shifted copy = cos
    n = 45  # 15 degrees * 3 ch
    assert torch.allclose(p15[..., 3 + n:6 + n], torch.sin(x + math.pi / 2),
                          atol=1e-6)


def test_attention_kv_swap_eager_semantics():
    """kv_swap on the eager path: batch b attends k/v of batch b^1."""
    import torch

    from novel_view_synthesis_3d_amd import ops
    torch.manual_seed(3)
    B, L, h, d = 4, 8, 2, 4
    q, k, v = (torch.randn(B, L, h, d) for _ in range(3))
    got = ops.attention(q, k, v, kv_swap=True)
    for b in range(B):
        want = ops.attention(q[b:b + 1], k[b ^ 1:(b ^ 1) + 1],
                             v[b ^ 1:(b ^ 1) + 1])
        assert torch.allclose(got[b], want[0], atol=1e-5), b


def test_linear_dispatch_cpu_matches_flinear():
    import torch
    import torch.nn.functional as F

    from novel_view_synthesis_3d_amd import ops
    torch.manual_seed(4)
    x = torch.randn(3, 7, 16, requires_grad=True)
    w = torch.randn(24, 16, requires_grad=True)
    b = torch.randn(24, requires_grad=True)
    y = ops.linear(x, w, b)
    assert torch.allclose(y, F.linear(x, w, b), atol=1e-6)
    y.sum().backward()
    x2 = x.detach().requires_grad_(True)
    w2 = w.detach().requires_grad_(True)
    b2 = b.detach().requires_grad_(True)
    F.linear(x2, w2, b2).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(w.grad, w2.grad, atol=1e-5)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)
