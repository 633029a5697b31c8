#!/usr/bin/env python3
"""Bisect the graph-replay NaN at full config: run the captured training
step under several ablations and report per-replay losses.

Run on a GPU box:  python tools/debug_graph.py
"""

import os
import sys

import torch

sys.path.insert(0, ".")


def run_variant(name, dropout, attn, env=None, steps=6, batch=4, seed=0):
    import importlib

    env = env or {}
    old = {}
    for k, v in env.items():
        old[k] = os.environ.get(k)
        os.environ[k] = v
    try:
        from novel_view_synthesis_3d_amd.config import (TrainConfig,
                                                        XUNetConfig)
        from novel_view_synthesis_3d_amd.engine.trainer import Trainer

        mcfg = XUNetConfig.full()
        mcfg.dropout = dropout
        if not attn:
            mcfg.attn_resolutions = ()
        tcfg = TrainConfig()
        tcfg.seed = seed
        tcfg.use_graph = os.environ.get("NVS3D_GRAPH_OFF") != "1"
        tcfg.amp = "bf16"
        tcfg.data = "synthetic"
        losses = []
        nosync = os.environ.get("NVS3D_DEBUG_NOSYNC") == "1"
        tr = Trainer(folder=None, train_batch_size=batch,
                     train_num_steps=steps, img_sidelength=128,
                     model_cfg=mcfg, train_cfg=tcfg)
        kept = []
        for i in range(steps):
            loss = tr.train_step()
            if nosync:
                kept.append(loss)  # queue replays back-to-back (bench.py style)
            else:
                torch.cuda.synchronize()
                losses.append(float(loss.item()))
        if nosync:
            torch.cuda.synchronize()
            losses = [float(l.item()) for l in kept[-3:]]
        print(f"[{name}] losses: "
              + " ".join(f"{v:.4f}" for v in losses), flush=True)
        del tr
        torch.cuda.empty_cache()
    except Exception as e:
        print(f"[{name}] FAILED: {e!r}", flush=True)
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def benchlike(tag="", env=None):
    """EXACTLY bench.py's structure: 3 unsynced warmup steps, one sync,
    8 unsynced timed steps, sync, read the loss once."""
    import time
    old = {}
    for k, v in (env or {}).items():
        old[k] = os.environ.get(k)
        os.environ[k] = v
    from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
    from novel_view_synthesis_3d_amd.engine.trainer import Trainer
    tcfg = TrainConfig()
    tcfg.data = "synthetic"
    tcfg.amp = "bf16"
    tcfg.seed = 1234
    tcfg.use_graph = True
    tr = Trainer(None, train_batch_size=16, img_sidelength=128,
                 train_num_steps=10 ** 9, model_cfg=XUNetConfig.named("full"),
                 train_cfg=tcfg)
    for _ in range(3):
        tr.train_step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(8):
        loss = tr.train_step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 8
    wn = float(tr.model.Conv_0.weight.norm().item())
    print(f"[benchlike{tag}] final loss {float(loss.item()):.4f} "
          f"wnorm {wn:.4f} ms/step {dt*1e3:.1f}", flush=True)
    del tr
    torch.cuda.empty_cache()
    for k, v in old.items():
        if v is None:
            os.environ.pop(k, None)
        else:
            os.environ[k] = v


def main():
    torch.manual_seed(0)
    import sys
    if len(sys.argv) > 1 and sys.argv[1] == "long":
        # reproduce bench.py's setup: seed 1234, 14 steps, b16
        run_variant("graph-b16-s1234-long", dropout=0.1, attn=True,
                    batch=16, steps=14, seed=1234)
        run_variant("eager-b16-s1234", dropout=0.1, attn=True,
                    batch=16, steps=14, seed=1234,
                    env={"NVS3D_GRAPH_OFF": "1"})
        run_variant("graph-b16-NOSYNC", dropout=0.1, attn=True,
                    batch=16, steps=14, seed=1234,
                    env={"NVS3D_DEBUG_NOSYNC": "1"})
        benchlike("-replaycap")  # new default: replay on capture stream
        benchlike("-replaycur", env={"NVS3D_GRAPH_REPLAY": "current"})
        benchlike("-capdefault", env={"NVS3D_GRAPH_CAPTURE": "default",
                                      "NVS3D_GRAPH_REPLAY": "current"})
        return
    if len(sys.argv) > 1 and sys.argv[1] == "batch":
        run_variant("graph-b8", dropout=0.1, attn=True, batch=8)
        run_variant("graph-b12", dropout=0.1, attn=True, batch=12)
        run_variant("graph-b16", dropout=0.1, attn=True, batch=16)
        run_variant("graph-b16-nodrop", dropout=0.0, attn=True, batch=16)
        run_variant("graph-b16-libpaths", dropout=0.1, attn=True, batch=16,
                    env={"NVS3D_ATTN_BWD": "gemm", "NVS3D_WGRAD": "im2col"})
        run_variant("graph-b16-noattn", dropout=0.1, attn=False, batch=16)
        return
    run_variant("baseline-graph", dropout=0.1, attn=True)
    run_variant("no-dropout", dropout=0.0, attn=True)
    run_variant("no-attn", dropout=0.1, attn=False)
    run_variant("attn-bwd-gemm", dropout=0.1, attn=True,
                env={"NVS3D_ATTN_BWD": "gemm"})
    run_variant("wgrad-im2col", dropout=0.1, attn=True,
                env={"NVS3D_WGRAD": "im2col"})


if __name__ == "__main__":
    main()
