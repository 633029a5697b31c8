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


def _load_flax_arrays(model, arrays: Dict[str, np.ndarray],
                      strict: bool) -> None:
    """Copy a {flax_path: ndarray} dict into the model (layout transforms
    from flax_tree). Handles two reference quirks:
      * pmap-stacked checkpoints (leading device axis): takes
        replica 0 when the array has exactly one extra leading dim;
      * bfloat16 arrays arrive as uint16 bit patterns (numpy has no bf16).
    """
    tree = flax_tree(model)
    missing = set(tree) - set(arrays)
    extra = set(arrays) - set(tree)
    if strict and (missing or extra):
        raise KeyError(f"flax tree mismatch: missing={sorted(missing)[:5]} "
                       f"extra={sorted(extra)[:5]}")
    with torch.no_grad():
        for k, (p, to_f, from_f) in tree.items():
            if k not in arrays:
                continue
            a = arrays[k]
            want = tuple(to_f(p.detach().cpu()).shape)
            if a.ndim == len(want) + 1 and tuple(a.shape[1:]) == want:
                a = a[0]  # pmap device axis (reference train.py:161-167)
            if a.dtype == np.uint16:  # bf16 bit pattern
                t = torch.from_numpy(np.ascontiguousarray(a)).view(
                    torch.bfloat16).float()
            else:
                t = torch.from_numpy(np.ascontiguousarray(a))
            p.copy_(from_f(t).to(p.dtype))


def import_flax_npz(model: torch.nn.Module, path: str, strict: bool = True) -> None:
    data = np.load(path)
    _load_flax_arrays(model, {k: data[k] for k in data.files}, strict)


# ---------------------------------------------------------------------------
# flax msgpack interchange — the reference's actual checkpoint format
# (flax.training.checkpoints.save_checkpoint writes
# flax.serialization.to_bytes(params) to `<dir>/<prefix><step>`:
# /root/reference/train.py:161-167, sampling.py:106-114). Wire format:
# msgpack maps of maps; ndarray leaves are ExtType 1 wrapping
# packb((shape, dtype.name, raw C-order bytes)) — flax/serialization.py.
# ---------------------------------------------------------------------------

_EXT_NDARRAY = 1


def _flax_ext_pack(x):
    import msgpack
    if isinstance(x, np.ndarray):
        payload = msgpack.packb((x.shape, x.dtype.name, x.tobytes("C")),
                                use_bin_type=True)
        return msgpack.ExtType(_EXT_NDARRAY, payload)
    raise TypeError(f"cannot pack {type(x)}")


def _flax_ext_unpack(code, data):
    import msgpack
    if code != _EXT_NDARRAY:
        return msgpack.ExtType(code, data)
    shape, dtype_name, buf = msgpack.unpackb(data, raw=True)
    if isinstance(dtype_name, bytes):
        dtype_name = dtype_name.decode()
    if dtype_name == "bfloat16":  # jax ml_dtypes name; keep the bits
        return np.frombuffer(buf, dtype=np.uint16).reshape(shape)
    return np.frombuffer(buf, dtype=np.dtype(dtype_name)).reshape(shape)


def _flatten(tree, prefix="", out=None):
    if out is None:
        out = {}
    for k, v in tree.items():
        key = k.decode() if isinstance(k, bytes) else str(k)
        if isinstance(v, dict):
            _flatten(v, prefix + key + "/", out)
        else:
            out[prefix + key] = v
    return out


def export_flax_msgpack(model: torch.nn.Module, path: str) -> None:
    """Write the param tree in the reference's flax msgpack format."""
    import msgpack
    nested: Dict = {}
    for k, (p, to_f, _) in flax_tree(model).items():
        node = nested
        parts = k.split("/")
        for part in parts[:-1]:
            node = node.setdefault(part, {})
        node[parts[-1]] = to_f(p.detach().cpu()).numpy()
    with open(path, "wb") as f:
        f.write(msgpack.packb(nested, default=_flax_ext_pack,
                              strict_types=True))


def import_flax_msgpack(model: torch.nn.Module, path: str,
                        strict: bool = True) -> None:
    """Load a flax msgpack checkpoint (as written by the reference trainer)
    into the model, transposing flax layouts and un-stacking pmap replicas."""
    import msgpack
    with open(path, "rb") as f:
        tree = msgpack.unpackb(f.read(), ext_hook=_flax_ext_unpack,
                               raw=False, strict_map_key=False)
    _load_flax_arrays(model, _flatten(tree), strict)


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
