"""FLAX-checkpoint import direction: build an npz with the exact tree naming
a converted reference checkpoint would have (flax kernel layouts) and verify
the model loads it and produces the mathematically corresponding output."""

import numpy as np
import torch

from novel_view_synthesis_3d_amd.config import XUNetConfig
from novel_view_synthesis_3d_amd.engine.checkpoint import (
    export_flax_npz, flax_tree, import_flax_npz,
)
from novel_view_synthesis_3d_amd.models.xunet import XUNet


def test_import_flax_layout_npz(tmp_path):
    """Simulate a converted reference checkpoint: arrays in FLAX layouts
    keyed by flax paths. Import must transpose them into our layouts."""
    torch.manual_seed(0)
    cfg = XUNetConfig(ch=8, ch_mult=(1, 2), emb_ch=8, num_res_blocks=1,
                      attn_resolutions=(8,), dropout=0.0)
    model = XUNet(cfg, img_sidelength=16)

    rng = np.random.default_rng(0)
    arrays = {}
    for path, (p, to_f, _) in flax_tree(model).items():
        flax_shape = tuple(to_f(p.detach()).shape)
        arrays[path] = rng.standard_normal(flax_shape).astype(np.float32) * 0.1
    npz = tmp_path / "converted.npz"
    np.savez(npz, **arrays)

    import_flax_npz(model, str(npz))
    # spot-check the transposes: conv kernel (1,3,3,Cin,Cout) -> (Cout,3,3,Cin)
    k = arrays["Conv_0/kernel"]  # (1,3,3,3,8)
    got = model.Conv_0.weight.detach().numpy()  # (8,3,3,3) OHWI
    np.testing.assert_allclose(got, np.transpose(k[0], (3, 0, 1, 2)),
                               rtol=1e-6)
    # Dense kernel (in,out) -> weight (out,in)
    kd = arrays["ConditioningProcessor_0/Dense_0/kernel"]
    gd = model.ConditioningProcessor_0.Dense_0.weight.detach().numpy()
    np.testing.assert_allclose(gd, kd.T, rtol=1e-6)

    # and exporting again reproduces the same flax arrays (roundtrip)
    out_npz = tmp_path / "roundtrip.npz"
    export_flax_npz(model, str(out_npz))
    data = np.load(out_npz)
    for k in arrays:
        np.testing.assert_allclose(data[k], arrays[k], rtol=1e-5,
                                   err_msg=k)


def test_strict_mismatch_raises(tmp_path):
    cfg = XUNetConfig(ch=8, ch_mult=(1, 2), emb_ch=8, num_res_blocks=1,
                      attn_resolutions=(8,), dropout=0.0)
    model = XUNet(cfg, img_sidelength=16)
    np.savez(tmp_path / "bad.npz", **{"not/a/param": np.zeros(3)})
    import pytest
    with pytest.raises(KeyError):
        import_flax_npz(model, str(tmp_path / "bad.npz"))


def _tiny_model():
    torch.manual_seed(1)
    cfg = XUNetConfig(ch=8, ch_mult=(1, 2), emb_ch=8, num_res_blocks=1,
                      attn_resolutions=(8,), dropout=0.0)
    return XUNet(cfg, img_sidelength=16)


def test_flax_msgpack_roundtrip(tmp_path):
    """export -> import through the flax msgpack wire format preserves every
    parameter bit-exactly (fp32)."""
    from novel_view_synthesis_3d_amd.engine.checkpoint import (
        export_flax_msgpack, import_flax_msgpack,
    )
    m1 = _tiny_model()
    path = tmp_path / "ckpt_msgpack"
    export_flax_msgpack(m1, str(path))

    m2 = _tiny_model()
    with torch.no_grad():
        for p in m2.parameters():
            p.add_(1.0)  # make them differ
    import_flax_msgpack(m2, str(path))
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(),
                                  m2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2), n1


def test_flax_msgpack_reference_style_file(tmp_path):
    """A checkpoint file built the way the REFERENCE writes it — raw msgpack
    map-of-maps with ExtType-1 ndarray leaves in flax layouts, pmap-stacked
    with a leading device axis (train.py:161-167 / D4) — imports correctly."""
    import msgpack

    from novel_view_synthesis_3d_amd.engine.checkpoint import (
        flax_tree, import_flax_msgpack,
    )

    model = _tiny_model()
    rng = np.random.default_rng(3)
    arrays = {}
    nested = {}
    for path_key, (p, to_f, _) in flax_tree(model).items():
        flax_shape = tuple(to_f(p.detach()).shape)
        a = rng.standard_normal((2,) + flax_shape).astype(np.float32) * 0.1
        arrays[path_key] = a  # device axis of 2 replicas
        node = nested
        parts = path_key.split("/")
        for part in parts[:-1]:
            node = node.setdefault(part, {})
        node[parts[-1]] = a

    def ext_pack(x):
        payload = msgpack.packb((x.shape, x.dtype.name, x.tobytes("C")),
                                use_bin_type=True)
        return msgpack.ExtType(1, payload)

    f = tmp_path / "model0"
    f.write_bytes(msgpack.packb(nested, default=ext_pack, strict_types=True))

    import_flax_msgpack(model, str(f))
    # replica 0 must have been taken and layout-transposed
    k = arrays["Conv_0/kernel"][0]  # (1,3,3,3,8) flax HWIO
    got = model.Conv_0.weight.detach().numpy()
    np.testing.assert_allclose(got, np.transpose(k[0], (3, 0, 1, 2)),
                               rtol=1e-6)


def test_flax_msgpack_bf16_leaves(tmp_path):
    """bfloat16 flax arrays (jax default dtype in some configs) arrive as
    uint16 bit patterns and are widened to the param dtype."""
    import msgpack

    from novel_view_synthesis_3d_amd.engine.checkpoint import (
        flax_tree, import_flax_msgpack,
    )
    model = _tiny_model()
    nested = {}
    want = {}
    for path_key, (p, to_f, _) in flax_tree(model).items():
        fa = to_f(p.detach().cpu())
        bf = fa.to(torch.bfloat16)
        want[path_key] = bf.float()
        u16 = bf.view(torch.uint16).numpy()
        node = nested
        parts = path_key.split("/")
        for part in parts[:-1]:
            node = node.setdefault(part, {})
        node[parts[-1]] = u16

    def ext_pack(x):
        # what flax writes for a bf16 array: dtype.name == 'bfloat16'
        payload = msgpack.packb((x.shape, "bfloat16", x.tobytes("C")),
                                use_bin_type=True)
        return msgpack.ExtType(1, payload)

    f = tmp_path / "model_bf16"
    f.write_bytes(msgpack.packb(nested, default=ext_pack, strict_types=True))
    import_flax_msgpack(model, str(f))
    tree = flax_tree(model)
    for path_key, (p, to_f, _) in tree.items():
        np.testing.assert_allclose(to_f(p.detach().cpu()).numpy(),
                                   want[path_key].numpy(), rtol=1e-6,
                                   err_msg=path_key)


def test_flax_msgpack_strict_mismatch_raises(tmp_path):
    import msgpack
    import pytest

    from novel_view_synthesis_3d_amd.engine.checkpoint import (
        import_flax_msgpack,
    )

    def ext_pack(x):
        payload = msgpack.packb((x.shape, x.dtype.name, x.tobytes("C")),
                                use_bin_type=True)
        return msgpack.ExtType(1, payload)

    model = _tiny_model()
    f = tmp_path / "bad_msgpack"
    f.write_bytes(msgpack.packb(
        {"not": {"a": {"param": np.zeros(3, dtype=np.float32)}}},
        default=ext_pack, strict_types=True))
    with pytest.raises(KeyError):
        import_flax_msgpack(model, str(f))
