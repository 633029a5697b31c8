#!/usr/bin/env python3
"""Attention backward microbench: fused flash bwd (attn_bwd.hip) vs the
GEMM-recompute path, on the model's hot shapes.

Run on a GPU box:  python tools/bench_attn_bwd.py > gpurun_out/attn_bwd_bench.jsonl
"""

import json
import os
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


def main():
    from novel_view_synthesis_3d_amd.ops import hip_ops
    shapes = [
        (16, 1024, 4, 128),   # full config res 32 (b16 per-frame batch)
        (32, 1024, 4, 128),
        (16, 256, 4, 256),    # res 16 (GEMM fallback)
        (16, 1024, 4, 16),    # small config
    ]
    for B, L, h, d in shapes:
        q, k, v = (torch.randn(B, L, h, d, device="cuda",
                               dtype=torch.bfloat16) for _ in range(3))
        do = torch.randn(B, L, h, d, device="cuda", dtype=torch.bfloat16)

        def run_bwd():
            qq = q.detach().requires_grad_(True)
            kk = k.detach().requires_grad_(True)
            vv = v.detach().requires_grad_(True)
            y = hip_ops.attention(qq, kk, vv)
            y.backward(do)
            return qq.grad, kk.grad, vv.grad

        os.environ["NVS3D_ATTN_BWD"] = "fused"
        t_fused = timeit(run_bwd)
        g_f = run_bwd()
        os.environ["NVS3D_ATTN_BWD"] = "gemm"
        t_gemm = timeit(run_bwd)
        g_g = run_bwd()
        os.environ["NVS3D_ATTN_BWD"] = "fused"

        rel = max(((a - b).abs().max() / (b.abs().max() + 1e-6)).item()
                  for a, b in zip(g_f, g_g))
        # 5 L^2 d GEMM-equivalents in the math (S, dP, dV, dK, dQ)
        flops = 5 * 2.0 * B * h * L * L * d
        rec = {"op": "attn_bwd", "shape": [B, L, h, d],
               "fused_ms": round(t_fused, 3), "gemm_ms": round(t_gemm, 3),
               "fused_tflops": round(flops / t_fused / 1e9, 1),
               "speedup": round(t_gemm / t_fused, 2),
               "rel_err_fused_vs_gemm": round(rel, 5)}
        print(json.dumps(rec), flush=True)


if __name__ == "__main__":
    main()
