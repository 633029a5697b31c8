"""SRN dataset tests: temp on-disk layout, parsing, collate (D1 fix)."""

import os

import numpy as np
import pytest
import torch

from novel_view_synthesis_3d_amd.data import io as data_io
from novel_view_synthesis_3d_amd.data.srn import SceneClassDataset
from novel_view_synthesis_3d_amd.data.synthetic import SyntheticSceneDataset


def make_srn_tree(root, n_instances=2, n_views=3, H=24):
    rng = np.random.default_rng(0)
    for i in range(n_instances):
        inst = os.path.join(root, f"inst{i:02d}")
        os.makedirs(os.path.join(inst, "rgb"))
        os.makedirs(os.path.join(inst, "pose"))
        with open(os.path.join(inst, "intrinsics.txt"), "w") as f:
            f.write("131.25 12.0 12.0 0.\n0. 0. 0.\n1.\n24 24\n")
        for v in range(n_views):
            img = rng.random((H, H, 3)).astype(np.float32)
            np.save(os.path.join(inst, "rgb", f"{v:06d}.npy"), img)
            pose = np.eye(4, dtype=np.float32)
            pose[:3, 3] = [0, 0, 1.3 + v * 0.1]
            if v % 2 == 0:  # one-line format
                with open(os.path.join(inst, "pose", f"{v:06d}.txt"), "w") as f:
                    f.write(" ".join(str(x) for x in pose.reshape(-1)))
            else:  # 4-line format
                with open(os.path.join(inst, "pose", f"{v:06d}.txt"), "w") as f:
                    for r in range(4):
                        f.write(" ".join(str(x) for x in pose[r]) + "\n")
    return root


def test_pose_both_formats(tmp_path):
    root = make_srn_tree(str(tmp_path))
    p0 = data_io.load_pose(os.path.join(root, "inst00", "pose", "000000.txt"))
    p1 = data_io.load_pose(os.path.join(root, "inst00", "pose", "000001.txt"))
    assert p0.shape == (4, 4) and p1.shape == (4, 4)
    assert np.allclose(p0[:3, :3], np.eye(3))
    assert np.allclose(p1[:3, 3], [0, 0, 1.4])


def test_parse_intrinsics_rescale(tmp_path):
    root = make_srn_tree(str(tmp_path))
    K, bary, scale, w2c = data_io.parse_intrinsics(
        os.path.join(root, "inst00", "intrinsics.txt"), trgt_sidelength=12)
    # f, cx, cy rescaled by 12/24 (reference util.py:64-67)
    assert K[0, 0] == pytest.approx(131.25 / 2)
    assert K[0, 2] == pytest.approx(6.0)
    assert K[1, 2] == pytest.approx(6.0)
    assert scale == 1.0 and w2c is False


def test_dataset_getitem_and_collate(tmp_path):
    root = make_srn_tree(str(tmp_path))
    ds = SceneClassDataset(root_dir=root, img_sidelength=24,
                           max_num_instances=-1,
                           max_observations_per_instance=50,
                           samples_per_instance=1)
    assert len(ds) == 6
    obs, gt = ds[0]
    assert isinstance(obs, list) and isinstance(gt, list)
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False,
                                     collate_fn=ds.collate_fn)
    batch_obs, batch_gt = next(iter(dl))
    # D1 fix: every field a stacked torch.Tensor
    for k, v in batch_obs.items():
        assert isinstance(v, torch.Tensor), k
        assert v.shape[0] == 2
    assert batch_obs["x"].shape == (2, 24, 24, 3)
    assert batch_obs["K"].shape == (2, 3, 3)
    assert batch_obs["x"].min() >= -1.0 and batch_obs["x"].max() <= 1.0


def test_dataset_cpu_noising_parity(tmp_path):
    """cpu_noising=True reproduces the reference worker-side noising
    (data_loader.py:92-110): z = sqrt(abar_t) x + sqrt(1-abar_t) eps."""
    from novel_view_synthesis_3d_amd.diffusion.schedules import DiffusionSchedule
    root = make_srn_tree(str(tmp_path))
    ds = SceneClassDataset(root_dir=root, img_sidelength=24,
                           samples_per_instance=1, cpu_noising=True)
    obs, gt = ds[0]
    s = obs[0]
    assert isinstance(s["z"], torch.Tensor)          # D1 fixed
    assert isinstance(s["logsnr"], torch.Tensor)
    sched = DiffusionSchedule(1000)
    t = int(s["t"])
    expect = (sched.sqrt_alphas_cumprod[t] * s["x_target"]
              + sched.sqrt_one_minus_alphas_cumprod[t] * s["noise"])
    assert torch.allclose(s["z"], expect, atol=1e-5)
    assert "noise" in gt[0]


def test_max_observations_subsample(tmp_path):
    root = make_srn_tree(str(tmp_path), n_views=3)
    ds = SceneClassDataset(root_dir=root, img_sidelength=24,
                           max_observations_per_instance=2,
                           samples_per_instance=1)
    assert len(ds) == 4  # 2 instances x 2 views


def test_synthetic_dataset():
    ds = SyntheticSceneDataset(num_instances=2, views_per_instance=3,
                               img_sidelength=16)
    assert len(ds) == 6
    s = ds[0]
    assert s["x"].shape == (16, 16, 3)
    batch = SyntheticSceneDataset.collate_fn([ds[0], ds[1]])
    assert batch["x"].shape == (2, 16, 16, 3)
    # deterministic per index
    s2 = ds[0]
    assert torch.equal(s["x"], s2["x"])
