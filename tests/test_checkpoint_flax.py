"""FLAX-compatible parameter layout tests (engine/checkpoint.py)."""

import numpy as np
import torch

from novel_view_synthesis_3d_amd.config import XUNetConfig
from novel_view_synthesis_3d_amd.engine.checkpoint import (
    export_flax_npz, flax_tree, import_flax_npz,
)
from novel_view_synthesis_3d_amd.models.xunet import XUNet


def small_model():
    torch.manual_seed(0)
    return XUNet(XUNetConfig.small(), img_sidelength=64)


def test_flax_tree_covers_all_params_once():
    model = small_model()
    tree = flax_tree(model)
    tree_ids = [id(p) for p, _, _ in tree.values()]
    model_ids = [id(p) for p in model.parameters()]
    assert sorted(tree_ids) == sorted(model_ids)
    assert len(set(tree.keys())) == len(tree)


def test_flax_naming_structure():
    model = small_model()
    keys = set(flax_tree(model).keys())
    # conditioning processor
    assert "ConditioningProcessor_0/Dense_0/kernel" in keys
    assert "ConditioningProcessor_0/Conv_0/kernel" in keys
    assert "ConditioningProcessor_0/Conv_1/kernel" in keys
    # stem + head convs at XUNet scope
    assert "Conv_0/kernel" in keys and "Conv_1/kernel" in keys
    # resblock internals, incl. the reference's nested GroupNorm wrapper
    assert "XUNetBlock_0/ResnetBlock_0/GroupNorm_0/GroupNorm_0/scale" in keys
    assert "XUNetBlock_0/ResnetBlock_0/Conv_0/kernel" in keys
    assert "XUNetBlock_0/ResnetBlock_0/FiLM_0/Dense_0/kernel" in keys
    # attention at res 32 (small config: XUNetBlock_2 is the first attn one)
    assert "XUNetBlock_2/AttnBlock_0/AttnLayer_0/DenseGeneral_0/kernel" in keys
    # end groupnorm
    assert "GroupNorm_0/GroupNorm_0/scale" in keys
    # down/up-resample resblocks exist at XUNet scope
    assert any(k.startswith("ResnetBlock_0/") for k in keys)


def test_flax_kernel_layouts():
    model = small_model()
    tree = flax_tree(model)
    p, to_f, _ = tree["Conv_0/kernel"]
    f = to_f(p)
    # flax conv kernel: (1, 3, 3, Cin, Cout) = (1,3,3,3,32)
    assert tuple(f.shape) == (1, 3, 3, 3, 32)
    p, to_f, _ = tree["ConditioningProcessor_0/Dense_0/kernel"]
    assert tuple(to_f(p).shape) == (32, 32)  # (in, out)
    p, to_f, _ = tree[
        "XUNetBlock_2/AttnBlock_0/AttnLayer_0/DenseGeneral_0/kernel"]
    f = to_f(p)
    assert f.ndim == 3 and f.shape[1] == 4  # (C, heads, head_dim)


def test_npz_roundtrip(tmp_path):
    model = small_model()
    path = str(tmp_path / "params.npz")
    export_flax_npz(model, path)

    model2 = XUNet(XUNetConfig.small(), img_sidelength=64)
    # perturb then restore
    with torch.no_grad():
        for p in model2.parameters():
            p.add_(1.0)
    import_flax_npz(model2, path)
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_npz_keys_are_flax_paths(tmp_path):
    model = small_model()
    path = str(tmp_path / "params.npz")
    export_flax_npz(model, path)
    data = np.load(path)
    assert "Conv_0/kernel" in data.files
    assert data["Conv_0/kernel"].shape == (1, 3, 3, 3, 32)
