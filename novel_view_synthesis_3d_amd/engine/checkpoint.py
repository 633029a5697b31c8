"""Checkpointing with a FLAX-compatible parameter layout.

Fixes the reference's gaps: the reference saves
pmap-stacked params only (train.py:161-167) and cannot resume (no optimizer
state, step resets). Here a checkpoint carries unreplicated params + Adam
m/v + step + config, and the parameter tree maps 1:1 onto the FLAX naming the
reference's `XUNet.init` produces (sampling.py:99-102), so reference
checkpoints are importable/exportable.

FLAX layout notes:
  * conv kernels: (1, 3, 3, Cin, Cout)  <-> ours (Cout, Cin, 3, 3)
  * Dense kernels: (in, out)            <-> ours (out, in)
  * DenseGeneral q/k/v: (C, heads, d)   <-> ours (heads*d, C)
  * the reference's GroupNorm wrapper nests flax nn.GroupNorm, so paths end
    in 'GroupNorm_k/GroupNorm_0/{scale,bias}'
Export/import uses .npz with '/'-joined keys (flax msgpack itself needs the
flax package, which this image doesn't ship; the tree *naming* is the
contract).
"""

from __future__ import annotations

import os
import re
from typing import Dict, Optional

import numpy as np
import torch


def _identity(x):
    return x


def flax_tree(model: torch.nn.Module) -> Dict[str, tuple]:
    """Walk the model, returning {flax_path: (param, to_flax, from_flax)}.

    Module attribute names ARE the flax submodule names by construction
    (models/layers.py, models/xunet.py).
    """
    out: Dict[str, tuple] = {}

    def walk(mod: torch.nn.Module, prefix: str) -> None:
        leaves = mod.flax_leaves() if hasattr(mod, "flax_leaves") else []
        leaf_params = [lp for _, lp, _, _ in leaves]
        for rel, p, to_f, from_f in leaves:
            out[prefix + rel] = (p, to_f, from_f)
        for name, p in mod.named_parameters(recurse=False):
            if not any(p is lp for lp in leaf_params):
                out[prefix + name] = (p, _identity, _identity)
        for name, child in mod.named_children():
            walk(child, prefix + name + "/")

    walk(model, "")
    return out


def export_flax_npz(model: torch.nn.Module, path: str) -> None:
    tree = flax_tree(model)
    arrays = {k: to_f(p.detach().cpu()).numpy() for k, (p, to_f, _) in tree.items()}
    np.savez(path, **arrays)


def import_flax_npz(model: torch.nn.Module, path: str, strict: bool = True) -> None:
    data = np.load(path)
    tree = flax_tree(model)
    missing = set(tree) - set(data.files)
    extra = set(data.files) - set(tree)
    if strict and (missing or extra):
        raise KeyError(f"flax tree mismatch: missing={sorted(missing)[:5]} "
                       f"extra={sorted(extra)[:5]}")
    with torch.no_grad():
        for k, (p, _, from_f) in tree.items():
            if k in data:
                p.copy_(from_f(torch.from_numpy(data[k])).to(p.dtype))


# ---------------------------------------------------------------------------
# Native checkpoints (torch format): params + optimizer + step + config
# ---------------------------------------------------------------------------

def save_checkpoint(ckpt_dir: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer], step: int,
                    extra: Optional[dict] = None, prefix: str = "model") -> str:
    os.makedirs(ckpt_dir, exist_ok=True)
    payload = {
        "format": "nvs3d-v1",
        "step": step,
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "extra": extra or {},
    }
    path = os.path.join(ckpt_dir, f"{prefix}_{step:08d}.pt")
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)
    latest = os.path.join(ckpt_dir, f"{prefix}_latest.txt")
    with open(latest, "w") as f:
        f.write(os.path.basename(path))
    return path


def find_latest(ckpt_dir: str, prefix: str = "model") -> Optional[str]:
    latest = os.path.join(ckpt_dir, f"{prefix}_latest.txt")
    if os.path.exists(latest):
        with open(latest) as f:
            name = f.read().strip()
        p = os.path.join(ckpt_dir, name)
        if os.path.exists(p):
            return p
    if not os.path.isdir(ckpt_dir):
        return None
    cands = [f for f in os.listdir(ckpt_dir)
             if re.fullmatch(rf"{prefix}_\d+\.pt", f)]
    if not cands:
        return None
    return os.path.join(ckpt_dir, sorted(cands)[-1])


def load_checkpoint(path: str, model: torch.nn.Module,
                    optimizer: Optional[torch.optim.Optimizer] = None,
                    map_location="cpu") -> int:
    payload = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(payload["model"])
    if optimizer is not None and payload.get("optimizer") is not None:
        optimizer.load_state_dict(payload["optimizer"])
    return int(payload["step"])
