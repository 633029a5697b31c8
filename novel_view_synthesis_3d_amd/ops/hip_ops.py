"""Python wrappers for the gfx950 HIP kernels (autograd.Function + dispatch).

Imports fail if the in-tree extension .so is missing — ops/__init__.py turns
that into a loud error on GPU boxes (no silent eager fallback).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from novel_view_synthesis_3d_amd.ops.build import built_path

_SO = built_path()
if not os.path.exists(_SO):
    raise ImportError(f"nvs3d_hip extension not built (expected {_SO})")
try:  # the build step may already have registered the library in-process
    torch.ops.nvs3d.gn_fwd
except (AttributeError, RuntimeError):
    torch.ops.load_library(_SO)
_OPS = torch.ops.nvs3d

# ops with a HIP implementation (consulted by ops/__init__.py dispatch)
HAS = {"joint_groupnorm", "pose_embedding", "frame_conv3x3", "attention",
       "nearest_upsample2x", "avgpool_downsample2x", "residual_scale_add",
       "linear"}


def conv_shapes_supported(cin: int, cout: int, stride: int) -> bool:
    """The implicit-GEMM kernel covers the FLOP-dominant convs (stride 1,
    Cin%64==0, Cout%128==0); stem (3ch), pose-emb (144ch), head (3ch) and
    small-config convs take the MIOpen path."""
    return stride == 1 and cin % 64 == 0 and cout % 128 == 0


# ---------------------------------------------------------------------------
# Fused joint-frame GroupNorm (+FiLM)(+SiLU)
# ---------------------------------------------------------------------------

class _GnFused(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, film, groups, eps, silu, p_drop, seed):
        x = x.contiguous()
        has_film = film is not None
        if has_film:
            film = film.to(x.dtype).contiguous()
        y, mean, rstd = _OPS.gn_fwd(x, gamma, beta, film, groups, eps, silu,
                                    p_drop, seed)
        extras = ([film] if has_film else []) + ([seed] if seed is not None
                                                 else [])
        ctx.save_for_backward(x, gamma, beta, mean, rstd, *extras)
        ctx.groups, ctx.silu, ctx.film = groups, silu, has_film
        ctx.p_drop, ctx.has_seed = p_drop, seed is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        saved = list(ctx.saved_tensors)
        seed = saved.pop() if ctx.has_seed else None
        film = saved.pop() if ctx.film else None
        x, gamma, beta, mean, rstd = saved
        outs = _OPS.gn_bwd(dy, x, gamma, beta, film, mean, rstd,
                           ctx.groups, ctx.silu, ctx.p_drop, seed)
        if ctx.film:
            dx, dgamma, dbeta, dfilm = outs
        else:
            dx, dgamma, dbeta = outs
            dfilm = None
        return (dx, dgamma.to(gamma.dtype), dbeta.to(beta.dtype),
                dfilm, None, None, None, None, None)


_GN_SEED_CTR: dict = {}


def _next_gn_seed(device: torch.device) -> torch.Tensor:
    """Per-call dropout seed SNAPSHOT from a device-resident counter.

    The clone + increment are device ops, so under hipGraph capture they are
    recorded and each replay advances the counter and re-snapshots — dropout
    masks keep changing across graph replays (the host-drawn-seed freeze the
    round-1 review flagged is gone). Counter init happens eagerly (warmup
    always precedes capture)."""
    idx = device.index if device.index is not None else 0
    ctr = _GN_SEED_CTR.get(idx)
    if ctr is None:
        init = int(torch.randint(0, 2 ** 30, (1,)).item())
        ctr = torch.full((1,), init, dtype=torch.int32, device=device)
        _GN_SEED_CTR[idx] = ctr
    seed = ctr.clone()
    # large odd stride decorrelates consecutive calls' hash streams
    ctr.add_(2654435761 & 0x7FFFFFFF)
    return seed


def joint_groupnorm(x, gamma, beta, groups, eps=1e-6, film=None, silu=False,
                    p_drop=0.0):
    """GroupNorm(+FiLM)(+SiLU)(+dropout) fully fused; the dropout mask is a
    counter-based hash regenerated in backward (no mask tensor)."""
    seed = _next_gn_seed(x.device) if p_drop > 0 else None
    return _GnFused.apply(x, gamma, beta, film, groups, eps, silu,
                          p_drop, seed)


# ---------------------------------------------------------------------------
# Fused rays + NeRF posenc + CFG mask (no grads w.r.t. any input)
# ---------------------------------------------------------------------------

def pose_embedding(R: torch.Tensor, t: torch.Tensor, K: torch.Tensor,
                   cond_mask: Optional[torch.Tensor], H: int, W: int,
                   out_dtype: torch.dtype) -> torch.Tensor:
    from novel_view_synthesis_3d_amd.models.rays import inv3x3
    Kinv = inv3x3(K)  # closed form: graph-capture-safe
    like = torch.empty(0, dtype=out_dtype, device=R.device)
    with torch.no_grad():
        return _OPS.rays_posenc(R.contiguous(), t.contiguous(), Kinv,
                                cond_mask, H, W, like)


# ---------------------------------------------------------------------------
# Fused multi-tensor Adam (used by engine/optim.py FusedAdam)
# ---------------------------------------------------------------------------

CHUNK = 1 << 16


class AdamPlan:
    """Cached pointer table + chunk map for a fixed set of (p,g,m,v)."""

    def __init__(self, tuples, device):
        self.key = [ (p.data_ptr(), g.data_ptr()) for p, g, _, _ in tuples ]
        ptrs, numels = [], []
        chunk_tensor, chunk_off = [], []
        for i, (p, g, m, v) in enumerate(tuples):
            assert p.dtype == torch.float32 and p.is_contiguous()
            ptrs += [p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr()]
            numels.append(p.numel())
            for off in range(0, p.numel(), CHUNK):
                chunk_tensor.append(i)
                chunk_off.append(off)
        self.ptrs = torch.tensor(ptrs, dtype=torch.uint64).to(device)
        self.numels = torch.tensor(numels, dtype=torch.int64).to(device)
        self.chunk_tensor = torch.tensor(chunk_tensor,
                                         dtype=torch.int32).to(device)
        self.chunk_off = torch.tensor(chunk_off, dtype=torch.int64).to(device)

    def matches(self, tuples) -> bool:
        return self.key == [(p.data_ptr(), g.data_ptr())
                            for p, g, _, _ in tuples]


def fused_adam_step(plan: AdamPlan, lr, b1, b2, eps, step: int) -> None:
    _OPS.fused_adam(plan.ptrs, plan.chunk_tensor, plan.chunk_off, plan.numels,
                    lr, b1, b2, eps, step)


# ---------------------------------------------------------------------------
# Implicit-GEMM 3x3 conv (K1): fwd + dgrad on the MFMA kernel, wgrad via
# MIOpen convolution_backward (library path for the long-K reduction GEMM).
# ---------------------------------------------------------------------------

class _FrameConv3x3(torch.autograd.Function):
    """MFMA igemm conv with an optionally fused residual tail:
    y = (conv(x, w) + bias + residual) * res_scale (the ResnetBlock ending,
    reference xunet.py:92)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, weight, bias, residual, res_scale):
        x = x.contiguous()
        weight = weight.contiguous()
        if residual is not None:
            residual = residual.contiguous()
        y = _OPS.conv3x3_fwd(x, weight, bias, residual, res_scale)
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        ctx.has_res = residual is not None
        ctx.res_scale = res_scale
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        if ctx.res_scale != 1.0:
            dy = dy * ctx.res_scale
        cout, _, _, cin = w.shape
        need_x, need_w, need_b = (ctx.needs_input_grad[0],
                                  ctx.needs_input_grad[1], ctx.has_bias)
        dx = dw = db = None
        if need_x:
            # dgrad = conv3x3(dy, w~), w~[ci,ey,ex,co] = w[co,2-ey,2-ex,ci]
            wt = torch.flip(w, dims=(1, 2)).permute(3, 1, 2, 0).contiguous()
            if conv_shapes_supported(cout, cin, 1):
                dx = _OPS.conv3x3_fwd(dy, wt, None, None, 1.0)
            else:
                dx = _miopen_conv(dy, wt, None)
        if need_w or need_b:
            dw4, db = _wgrad(x, w, dy, ctx.has_bias)
            if need_w:
                dw = dw4
        dres = dy if ctx.has_res else None
        return dx, dw, db, dres, None


def _nchw_view(x5):
    """(IMG,H,W,C) or (B,F,H,W,C) contiguous -> NCHW channels_last view."""
    if x5.dim() == 5:
        B, F, H, W, C = x5.shape
        x5 = x5.reshape(B * F, H, W, C)
    return x5.permute(0, 3, 1, 2)


def _miopen_conv(x, w, bias):
    import torch.nn.functional as Fn
    shape = x.shape
    y = Fn.conv2d(_nchw_view(x), w.permute(0, 3, 1, 2), bias, padding=1)
    out = y.permute(0, 2, 3, 1)
    return out.reshape(*shape[:-1], w.shape[0]).contiguous()


def _wgrad_shapes_supported(x, cout) -> bool:
    """The direct MFMA wgrad kernel covers W%16==0 and (W<=128 or W%128==0),
    Cin%64==0, Cout%128==0 — every conv the MFMA fwd kernel takes."""
    W = x.shape[-2]
    cin = x.shape[-1]
    return (cin % 64 == 0 and cout % 128 == 0 and W % 16 == 0
            and (W <= 128 or W % 128 == 0)
            and os.environ.get("NVS3D_WGRAD", "mfma") != "im2col")


def _wgrad(x, w, dy, has_bias):
    """Conv wgrad dispatch: hand-written direct MFMA kernel (one pass over
    x/dy, all 9 taps concurrent, fp32 split-K atomics — conv3x3_wgrad.hip),
    falling back to the 3-row-shift im2col+hipBLASLt decomposition for
    unsupported shapes."""
    cout = w.shape[0]
    if _wgrad_shapes_supported(x, cout):
        outs = _OPS.conv3x3_wgrad(x, dy, has_bias)
        dw = outs[0].to(x.dtype)
        db = outs[1].to(dy.dtype) if has_bias else None
        return dw, db
    return _wgrad_im2col_gemm(x, w, dy, has_bias)


def _wgrad_im2col_gemm(x, w, dy, has_bias):
    """Conv wgrad via the 3-row-shift decomposition: materialize only the
    ROW-shifted im2col A3[m][(dy,ci)] (3x the input instead of 9x); each
    kernel COLUMN dx is then one hipBLASLt GEMM between flat-shifted views:

        dW[:, dx] = A3[m-(dx-1)]^T @ dyf[m]  -  wrap corrections

    The flat +-1 shift wraps across image-row boundaries exactly at the
    rows with w==0 (dx=0) / w==W-1 (dx=2); the true contribution there is
    zero (SAME padding), so those rows' outer products (M/W rows) are
    gathered and subtracted — exact, tested against the fp32 oracle."""
    cout, _, _, cin = w.shape
    M = dy.numel() // cout
    W = x.shape[-2]
    dyf = dy.reshape(M, cout)
    A3 = _OPS.im2col3x3(x, 1, 3, 0, -1, None)        # (M, 3*Cin)
    dev = x.device

    blocks = []
    # dx = 0: pair A3[m-1] with dyf[m]
    g0 = torch.matmul(A3[:M - 1].transpose(0, 1), dyf[1:]).float()
    idx0 = torch.arange(W, M, W, device=dev)          # w==0, m>=1
    g0 -= torch.matmul(A3.index_select(0, idx0 - 1).transpose(0, 1),
                       dyf.index_select(0, idx0)).float()
    blocks.append(g0)
    # dx = 1: aligned
    blocks.append(torch.matmul(A3.transpose(0, 1), dyf).float())
    # dx = 2: pair A3[m+1] with dyf[m]
    g2 = torch.matmul(A3[1:].transpose(0, 1), dyf[:M - 1]).float()
    idx2 = torch.arange(W - 1, M - 1, W, device=dev)  # w==W-1, m<M-1
    g2 -= torch.matmul(A3.index_select(0, idx2 + 1).transpose(0, 1),
                       dyf.index_select(0, idx2)).float()
    blocks.append(g2)

    # blocks[dx]: (3*Cin, Cout), rows = (dy, ci) -> dW OHWI (Cout,3,3,Cin)
    dwf = torch.stack(blocks, dim=1)                  # (3*Cin, 3dx, Cout)
    dw = (dwf.reshape(3, cin, 3, cout).permute(3, 0, 2, 1)
          .contiguous().to(x.dtype))
    db = dyf.sum(dim=0, dtype=torch.float32).to(dy.dtype) if has_bias else None
    return dw, db


def _pad_c8(t):
    """Pad the channel (last) dim to a multiple of 8 (im2col needs 16B packs)."""
    c = t.shape[-1]
    pad = (-c) % 8
    if pad == 0:
        return t, c
    import torch.nn.functional as Fn
    return Fn.pad(t, (0, pad)), c


def _gemm_conv_fwd(x, w, bias, stride):
    """Generic 3x3 SAME conv for shapes the MFMA igemm kernel doesn't take
    (stem 3ch, pose-emb 144ch strided, head 3ch, small configs): strided
    im2col kernel + ONE hipBLASLt GEMM. Replaces MIOpen entirely."""
    cout, _, _, cin = w.shape
    xp, _ = _pad_c8(x.contiguous())
    A = _OPS.im2col3x3(xp, stride, 9, 0, -1, None)          # (Mo, 9*Cpad)
    cpad = xp.shape[-1]
    w2 = w.permute(1, 2, 3, 0)                            # (3,3,Cin,Cout)
    if cpad != cin:
        import torch.nn.functional as Fn
        w2 = Fn.pad(w2, (0, 0, 0, cpad - cin))
    wk = w2.reshape(9 * cpad, cout)
    y = torch.matmul(A, wk.to(A.dtype))
    if bias is not None:
        y = y + bias.to(y.dtype)
    H, W = x.shape[-3], x.shape[-2]
    Ho, Wo = -(-H // stride), -(-W // stride)
    return y.reshape(*x.shape[:-3], Ho, Wo, cout)


class _FrameConvGeneric(torch.autograd.Function):
    """Fallback conv shapes; fwd/wgrad via im2col+GEMM, dgrad via im2col of
    dy (stride 1 only — the model's strided convs consume no-grad inputs)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, weight, bias, stride):
        x = x.contiguous()
        weight = weight.contiguous()
        y = _gemm_conv_fwd(x, weight, bias, stride)
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        ctx.stride = stride
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        cout, _, _, cin = w.shape
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            assert ctx.stride == 1, "strided dgrad not needed by the model"
            wt = torch.flip(w, dims=(1, 2)).permute(3, 1, 2, 0).contiguous()
            if conv_shapes_supported(cout, cin, 1):
                dx = _OPS.conv3x3_fwd(dy, wt, None, None, 1.0)
            else:
                dx = _gemm_conv_fwd(dy, wt, None, 1)
        if ctx.needs_input_grad[1] or ctx.has_bias:
            dyf = dy.reshape(-1, cout)
            xp, _ = _pad_c8(x)
            cpad = xp.shape[-1]
            A = _OPS.im2col3x3(xp, ctx.stride, 9, 0, -1, None)
            dwf = torch.matmul(A.transpose(0, 1), dyf)    # (9*Cpad, Cout)
            dw = dwf.reshape(3, 3, cpad, cout)[:, :, :cin, :]                 .permute(3, 0, 1, 2).contiguous()
            if not ctx.needs_input_grad[1]:
                dw = None
            db = (dyf.sum(0, dtype=torch.float32).to(dy.dtype)
                  if ctx.has_bias else None)
        return dx, dw, db, None


def frame_conv3x3(x, weight, bias, stride: int = 1):
    cout, _, _, cin = weight.shape
    if conv_shapes_supported(cin, cout, stride):
        return _FrameConv3x3.apply(x, weight, bias, None, 1.0)
    if stride == 1 and cout % 128 == 0 and cin >= 64:
        # e.g. the 144ch stride-1 pose conv: zero-pad channels to %64 and
        # take the MFMA kernel (F.pad is differentiable; the pad columns
        # multiply zero weights). Far cheaper than a 9x im2col of the input.
        import torch.nn.functional as Fn
        pad = (-cin) % 64
        xp = Fn.pad(x, (0, pad))
        wp = Fn.pad(weight, (0, pad))
        return _FrameConv3x3.apply(xp, wp, bias, None, 1.0)
    return _FrameConvGeneric.apply(x, weight, bias, stride)


# ---------------------------------------------------------------------------
# Flash MFMA attention (K7): HIP forward (saves logsumexp), GEMM-recompute
# backward (rocBLAS batched matmuls — the "plain library GEMM" path).
# ---------------------------------------------------------------------------

def _swap_frame_pairs(t):
    """(2B', ...) -> pairs (2i, 2i+1) exchanged (cross-frame kv order)."""
    B2 = t.shape[0]
    return (t.reshape(B2 // 2, 2, *t.shape[1:]).flip(1)
            .reshape(t.shape).contiguous())


class _Attention(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, q, k, v, kv_swap):
        # strided views (fused-QKV slices) are consumed directly; the
        # kernel requires only a contiguous (head, d) tail
        if q.stride(-1) != 1 or q.stride(2) != q.size(3):
            q = q.contiguous()
        if k.stride(-1) != 1 or k.stride(2) != k.size(3):
            k = k.contiguous()
        if v.stride(-1) != 1 or v.stride(2) != v.size(3):
            v = v.contiguous()
        out, lse = _OPS.attn_fwd(q, k, v, kv_swap)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.kv_swap = kv_swap
        return out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        B, L, H, D = q.shape
        Lk = k.shape[1]
        swap = ctx.kv_swap
        if (D in (16, 32, 64, 128) and L % 128 == 0 and Lk % 128 == 0
                and os.environ.get("NVS3D_ATTN_BWD", "fused") != "gemm"):
            # fused flash-style backward (attn_bwd.hip): tile-wise recompute
            # of S/P from q/k/lse — no (B,H,L,L) tensor touches HBM
            doc = do.to(q.dtype).contiguous()
            delta = _OPS.attn_delta(doc, o)
            dq, dk, dv = _OPS.attn_bwd_fused(q, k, v, doc, lse, delta, swap)
            return dq, dk, dv, None
        if swap:
            # GEMM-recompute fallback shapes (d=256 / short L): materialize
            # the frame swap, then un-swap the kv grads
            k = _swap_frame_pairs(k)
            v = _swap_frame_pairs(v)
        scale = 1.0 / (D ** 0.5)
        # (B,H,L,D) strided views; rocBLAS consumes them without copies
        qt = q.permute(0, 2, 1, 3)
        kt = k.permute(0, 2, 1, 3)
        vt = v.permute(0, 2, 1, 3)
        dob = do.to(q.dtype).permute(0, 2, 1, 3)
        # recompute P in ONE pass (bf16 GEMM + fused exp(S*scale - lse))
        s = torch.matmul(qt, kt.transpose(-1, -2)).contiguous()
        p = _OPS.attn_p_from_lse(s, lse, scale)        # (B,H,L,Lk) bf16
        dv = torch.matmul(p.transpose(-1, -2), dob)
        dp = torch.matmul(dob, vt.transpose(-1, -2)).contiguous()
        delta = _OPS.attn_delta(do.to(q.dtype).contiguous(), o)
        ds = _OPS.attn_ds(p, dp, delta.permute(0, 2, 1).contiguous(), scale)
        dq = torch.matmul(ds, kt)
        dk = torch.matmul(ds.transpose(-1, -2), qt)
        dq = dq.permute(0, 2, 1, 3).contiguous()
        dk = dk.permute(0, 2, 1, 3).contiguous()
        dv = dv.permute(0, 2, 1, 3).contiguous()
        if swap:
            dk = _swap_frame_pairs(dk)
            dv = _swap_frame_pairs(dv)
        return dq, dk, dv, None


def attention(q, k, v, kv_swap: bool = False):
    return _Attention.apply(q, k, v, kv_swap)


# ---------------------------------------------------------------------------
# Linear (K6): hipBLASLt forward; custom split-K MFMA wgrad for the
# tall-skinny shapes hipBLASLt collapses on (gemm_wgrad.hip).
# ---------------------------------------------------------------------------

def _linear_wgrad_supported(M: int, N: int, K: int) -> bool:
    # measured crossover vs hipBLASLt: the split-K kernel wins decisively at
    # M >= ~1e5 (2.5-3.4x), ties at 32k, loses below (gpurun_out/dense_bench4)
    return (M >= 65536 and N % 8 == 0 and K % 8 == 0
            and os.environ.get("NVS3D_LINEAR_WGRAD", "mfma") != "blas")


class _LinearFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, w, b):
        y = torch.nn.functional.linear(x, w, b)
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        N, K = w.shape
        dy = dy.to(x.dtype)
        dyf = dy.reshape(-1, N).contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # ONE flat GEMM (a leading-dim batched matmul would shatter
            # into thousands of tiny per-row GEMMs on 4/5-D activations)
            dx = torch.matmul(dyf, w).view(x.shape)
        if ctx.needs_input_grad[1] or ctx.has_bias:
            xf = x.reshape(-1, K).contiguous()
            M = dyf.shape[0]
            if _linear_wgrad_supported(M, N, K):
                outs = _OPS.linear_wgrad(dyf, xf, ctx.has_bias)
                dw = outs[0].to(x.dtype)
                db = outs[1].to(dy.dtype) if ctx.has_bias else None
            else:
                dw = torch.matmul(dyf.transpose(0, 1), xf)
                db = (dyf.sum(0, dtype=torch.float32).to(dy.dtype)
                      if ctx.has_bias else None)
            if not ctx.needs_input_grad[1]:
                dw = None
        return dx, dw, db


def linear(x, w, b=None):
    return _LinearFn.apply(x, w, b)


def frame_conv3x3_residual(x, weight, bias, residual, res_scale):
    """y = (conv3x3(x,w) + bias + residual) * res_scale, residual fused in
    the conv epilogue when the MFMA kernel covers the shape."""
    cout, _, _, cin = weight.shape
    if conv_shapes_supported(cin, cout, 1):
        return _FrameConv3x3.apply(x, weight, bias, residual, res_scale)
    y = frame_conv3x3(x, weight, bias, 1)
    return (y + residual) * res_scale


# ---------------------------------------------------------------------------
# Resampling (K8/K9) + residual (K10) elementwise kernels
# ---------------------------------------------------------------------------

class _Up2x(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return _OPS.up2x_fwd(x.contiguous())

    @staticmethod
    def backward(ctx, dy):
        return _OPS.up2x_bwd(dy.contiguous())


class _Pool2x(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return _OPS.pool2x_fwd(x.contiguous())

    @staticmethod
    def backward(ctx, dy):
        return _OPS.pool2x_bwd(dy.contiguous())


class _AddScale(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, scale):
        ctx.scale = scale
        return _OPS.add_scale(a.contiguous(), b.contiguous(), scale)

    @staticmethod
    def backward(ctx, dy):
        g = dy * ctx.scale
        return g, g, None


def _elts_ok(x):
    return x.size(-1) % 8 == 0 and x.dtype in (torch.bfloat16, torch.float32)


def nearest_upsample2x(x):
    if _elts_ok(x):
        return _Up2x.apply(x)
    from novel_view_synthesis_3d_amd.ops import reference as _ref
    return _ref.nearest_upsample2x(x)


def avgpool_downsample2x(x):
    if _elts_ok(x):
        return _Pool2x.apply(x)
    from novel_view_synthesis_3d_amd.ops import reference as _ref
    return _ref.avgpool_downsample2x(x)


def residual_scale_add(h, h_in):
    import math
    if _elts_ok(h) and h.dtype == h_in.dtype:
        return _AddScale.apply(h, h_in, 1.0 / math.sqrt(2.0))
    from novel_view_synthesis_3d_amd.ops import reference as _ref
    return _ref.residual_scale_add(h, h_in)
