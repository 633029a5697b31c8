#!/usr/bin/env python3
"""Per-kernel microbenchmarks on MI355X: HIP kernels vs library/eager paths.

Run on a GPU box:  python tools/kernel_bench.py > gpurun_out/kernel_bench.json
"""

import json
import sys
import time

import torch

sys.path.insert(0, ".")


def timeit(fn, warmup=5, iters=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000  # ms


def bench_conv():
    from novel_view_synthesis_3d_amd.ops import hip_ops
    from novel_view_synthesis_3d_amd.ops import reference as ref
    # (B,F,H,W,Cin,Cout) — the full-config hot shapes
    shapes = [
        (8, 2, 128, 128, 256, 256),
        (8, 2, 64, 64, 512, 512),
        (8, 2, 32, 32, 512, 512),
        (8, 2, 16, 16, 1024, 1024),
        (8, 2, 16, 16, 2048, 1024),  # up-path concat
    ]
    out = []
    for B, F, H, W, Ci, Co in shapes:
        x = torch.randn(B, F, H, W, Ci, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(Co, 3, 3, Ci, device="cuda",
                        dtype=torch.bfloat16) * 0.02
        b = torch.randn(Co, device="cuda")
        flops = 2.0 * B * F * H * W * 9 * Ci * Co
        t_hip = timeit(
            lambda: torch.ops.nvs3d.conv3x3_fwd(x, w, b, None, 1.0))
        t_mio = timeit(lambda: ref.frame_conv3x3(x, w, b.to(torch.bfloat16)))
        rec = {"op": "conv3x3_fwd", "shape": [B, F, H, W, Ci, Co],
               "hip_ms": round(t_hip, 3), "miopen_ms": round(t_mio, 3),
               "hip_tflops": round(flops / t_hip / 1e9, 1),
               "miopen_tflops": round(flops / t_mio / 1e9, 1)}
        out.append(rec)
        print(json.dumps(rec), flush=True)
    return out


def bench_attn():
    from novel_view_synthesis_3d_amd.ops import reference as ref
    shapes = [
        (8, 1024, 4, 128),
        (8, 256, 4, 256),
        (16, 1024, 4, 128),
        (8, 1024, 4, 16),
    ]
    for B, L, h, d in shapes:
        q, k, v = (torch.randn(B, L, h, d, device="cuda",
                               dtype=torch.bfloat16) for _ in range(3))
        flops = 2.0 * B * h * L * L * d * 2
        t_hip = timeit(lambda: torch.ops.nvs3d.attn_fwd(q, k, v))
        t_ref = timeit(lambda: ref.attention(q, k, v))
        rec = {"op": "attn_fwd", "shape": [B, L, h, d],
               "hip_ms": round(t_hip, 3), "eager_ms": round(t_ref, 3),
               "hip_tflops": round(flops / t_hip / 1e9, 1),
               "eager_tflops": round(flops / t_ref / 1e9, 1)}
        print(json.dumps(rec), flush=True)


def bench_gn():
    from novel_view_synthesis_3d_amd.ops import hip_ops
    shapes = [(8, 2, 128, 128, 256), (8, 2, 64, 64, 512),
              (8, 2, 16, 16, 1024)]
    for B, F, H, W, C in shapes:
        x = torch.randn(B, F, H, W, C, device="cuda", dtype=torch.bfloat16)
        gm = torch.ones(C, device="cuda")
        bt = torch.zeros(C, device="cuda")
        fs = torch.randn_like(x) * 0.1
        ft = torch.randn_like(x) * 0.1
        nbytes = x.numel() * 2
        film = torch.cat([fs, ft], dim=-1).contiguous()
        t = timeit(lambda: torch.ops.nvs3d.gn_fwd(x, gm, bt, film, 32,
                                                  1e-6, True))
        # fwd traffic: read x twice + fs + ft + write y = 5 passes
        rec = {"op": "gn_fwd_film_silu", "shape": [B, F, H, W, C],
               "ms": round(t, 3),
               "eff_tb_s": round(5 * nbytes / t / 1e9, 2)}
        print(json.dumps(rec), flush=True)


def bench_pose():
    from novel_view_synthesis_3d_amd.ops import hip_ops
    from novel_view_synthesis_3d_amd.data.synthetic import random_cameras
    B, H = 8, 128
    g = torch.Generator(device="cuda").manual_seed(0)
    R1, t1, K = random_cameras(B, H, "cuda", g)
    R2, t2, _ = random_cameras(B, H, "cuda", g)
    R, t = torch.stack([R1, R2], 1), torch.stack([t1, t2], 1)
    m = torch.ones(B, device="cuda")
    tms = timeit(lambda: hip_ops.pose_embedding(R, t, K, m, H, H,
                                                torch.bfloat16))
    print(json.dumps({"op": "pose_embedding", "shape": [B, 2, H, H, 144],
                      "ms": round(tms, 3)}), flush=True)


if __name__ == "__main__":
    assert torch.cuda.is_available()
    bench_conv()
    bench_attn()
    bench_gn()
    bench_pose()
