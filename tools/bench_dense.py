#!/usr/bin/env python3
"""Dense-family GEMM shapes (FiLM, QKV, skip Dense) through hipBLASLt:
achieved TF + effective bandwidth vs the streaming roofline. Decides whether
a custom skinny-K kernel is worth writing.

Run on a GPU box:  python tools/bench_dense.py > gpurun_out/dense_bench.jsonl
"""

import json
import sys
import time

import torch

sys.path.insert(0, ".")


def timeit(fn, warmup=10, iters=50):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    from novel_view_synthesis_3d_amd.ops import hip_ops  # loads torch.ops.nvs3d
    # (tag, M, K, N) — b16 full-config shapes
    shapes = [
        ("film_fwd_L0", 16 * 2 * 128 * 128, 256, 512),
        ("film_fwd_L1", 16 * 2 * 64 * 64, 256, 1024),
        ("film_fwd_L3", 16 * 2 * 16 * 16, 256, 2048),
        ("film_dgrad_L0", 16 * 2 * 128 * 128, 512, 256),
        ("qkv_fwd_res32", 16 * 2 * 1024, 512, 1536),
        ("qkv_fwd_res16", 16 * 2 * 256, 1024, 3072),
        ("skip_dense_L0", 16 * 2 * 128 * 128, 256, 256),
    ]
    for tag, M, K, N in shapes:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t_lin = timeit(lambda: torch.nn.functional.linear(a, w))
        flops = 2.0 * M * N * K
        bytes_min = 2.0 * (M * K + M * N + N * K)
        rec = {"op": "dense", "tag": tag, "mnk": [M, N, K],
               "ms": round(t_lin, 4),
               "tflops": round(flops / t_lin / 1e9, 1),
               "eff_tbps": round(bytes_min / t_lin / 1e9, 2),
               "roofline_ms_at_6tbps": round(bytes_min / 6.0e12 * 1e3, 4)}
        print(json.dumps(rec), flush=True)
        # wgrad shape: dW = dy^T a  (K_gemm = M)
        dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        t_wg = timeit(lambda: torch.matmul(dy.transpose(0, 1), a))
        t_k = timeit(lambda: torch.ops.nvs3d.linear_wgrad(dy, a, True))
        dw_k = torch.ops.nvs3d.linear_wgrad(dy, a, True)[0]
        dw_b = torch.matmul(dy.transpose(0, 1).float(), a.float())
        rel = ((dw_k - dw_b).abs().max() / (dw_b.abs().max() + 1e-6)).item()
        rec = {"op": "dense_wgrad", "tag": tag + "_wgrad",
               "mnk": [N, K, M], "blas_ms": round(t_wg, 4),
               "mfma_ms": round(t_k, 4),
               "blas_tflops": round(flops / t_wg / 1e9, 1),
               "mfma_tflops": round(flops / t_k / 1e9, 1),
               "speedup": round(t_wg / t_k, 2),
               "rel_err": round(rel, 5)}
        print(json.dumps(rec), flush=True)
        del a, w, dy
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
