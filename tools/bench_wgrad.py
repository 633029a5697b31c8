#!/usr/bin/env python3
"""Wgrad microbench: direct MFMA kernel vs the 3-row-shift im2col+hipBLASLt
path, on the full-config hot shapes.

Run on a GPU box:  python tools/bench_wgrad.py > gpurun_out/wgrad_bench.jsonl
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


def main():
    from novel_view_synthesis_3d_amd.ops import hip_ops
    # (B,F,H,W,Cin,Cout) — full-config (b16) hot conv shapes
    shapes = [
        (16, 2, 128, 128, 256, 256),
        (16, 2, 64, 64, 512, 512),
        (16, 2, 32, 32, 512, 512),
        (16, 2, 16, 16, 1024, 1024),
        (16, 2, 16, 16, 2048, 1024),  # up-path concat
    ]
    for B, F, H, W, Ci, Co in shapes:
        x = torch.randn(B, F, H, W, Ci, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(Co, 3, 3, Ci, device="cuda",
                        dtype=torch.bfloat16) * 0.02
        dy = torch.randn(B, F, H, W, Co, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * B * F * H * W * 9 * Ci * Co

        t_mfma = timeit(lambda: torch.ops.nvs3d.conv3x3_wgrad(x, dy, True))
        t_i2c = timeit(lambda: hip_ops._wgrad_im2col_gemm(x, w, dy, True))

        # numerics cross-check (vs the already-parity-tested im2col path)
        dw_m = torch.ops.nvs3d.conv3x3_wgrad(x, dy, True)[0]
        dw_i = hip_ops._wgrad_im2col_gemm(x, w, dy, True)[0].float()
        rel = ((dw_m - dw_i).abs().max() /
               (dw_i.abs().max() + 1e-6)).item()

        rec = {"op": "conv3x3_wgrad", "shape": [B, F, H, W, Ci, Co],
               "mfma_ms": round(t_mfma, 3), "im2col_ms": round(t_i2c, 3),
               "mfma_tflops": round(flops / t_mfma / 1e9, 1),
               "im2col_tflops": round(flops / t_i2c / 1e9, 1),
               "speedup": round(t_i2c / t_mfma, 2),
               "rel_err_vs_im2col": round(rel, 5)}
        print(json.dumps(rec), flush=True)


if __name__ == "__main__":
    main()
