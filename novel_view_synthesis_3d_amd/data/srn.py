"""SRN/ShapeNet-style scene dataset.

Re-implements /root/reference/dataset/data_loader.py:27-196 with the defects
fixed and the forward-noising moved on-device:

* D1 fixed: every batch field collates to a torch.Tensor (the reference left
  `z` a list of numpy arrays and `logsnr` a numpy scalar, which crashes
  train.py:133).
* K19 moved: by default a sample carries the CLEAN target view `x_target`;
  the trainer noisifies the whole batch on the GPU (diffusion/forward.py).
  `cpu_noising=True` restores the reference behavior (z/noise/logsnr computed
  per-sample in the worker) for parity testing.

Disk layout (reference data_loader.py:40-50, util.py:46-81):
  root/<instance>/{rgb/*.png|*.npy, pose/*.txt, intrinsics.txt[, params/]}
"""

from __future__ import annotations

import os
from glob import glob
from typing import Dict, List, Optional

import numpy as np
import torch

from novel_view_synthesis_3d_amd.data import io as data_io
from novel_view_synthesis_3d_amd.diffusion.schedules import (
    DiffusionSchedule, logsnr_schedule_cosine,
)

_SCHEDULE = None


def _schedule() -> DiffusionSchedule:
    global _SCHEDULE
    if _SCHEDULE is None:
        _SCHEDULE = DiffusionSchedule(1000)
    return _SCHEDULE


def _pick(lst, idcs):
    if not lst:
        return lst
    return [lst[i] for i in idcs]


class SceneInstanceDataset:
    """All observations of one object instance
    (reference data_loader.py:27-113)."""

    def __init__(self, instance_idx: int, instance_dir: str,
                 specific_observation_idcs=None, img_sidelength=None,
                 num_images: int = -1, cpu_noising: bool = False):
        self.instance_idx = instance_idx
        self.instance_dir = instance_dir
        self.img_sidelength = img_sidelength
        self.cpu_noising = cpu_noising

        color_dir = os.path.join(instance_dir, "rgb")
        pose_dir = os.path.join(instance_dir, "pose")
        if not os.path.isdir(color_dir):
            raise FileNotFoundError(f"no rgb/ dir under {instance_dir}")
        self.color_paths = sorted(data_io.glob_imgs(color_dir))
        self.pose_paths = sorted(glob(os.path.join(pose_dir, "*.txt")))

        if specific_observation_idcs is not None:
            self.color_paths = _pick(self.color_paths, specific_observation_idcs)
            self.pose_paths = _pick(self.pose_paths, specific_observation_idcs)
        elif num_images != -1:
            # (fix) cap at the available view count: the reference's linspace
            # oversamples with duplicates when num_images > len
            num_images = min(num_images, len(self.color_paths))
            idcs = np.linspace(0, stop=len(self.color_paths), num=num_images,
                               endpoint=False, dtype=int)
            self.color_paths = _pick(self.color_paths, idcs)
            self.pose_paths = _pick(self.pose_paths, idcs)

    def set_img_sidelength(self, s: int) -> None:
        self.img_sidelength = s

    def __len__(self) -> int:
        return len(self.pose_paths)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        K4, _, _, _ = data_io.parse_intrinsics(
            os.path.join(self.instance_dir, "intrinsics.txt"),
            trgt_sidelength=self.img_sidelength)
        rgb = data_io.load_rgb(self.color_paths[idx], self.img_sidelength)
        pose = data_io.load_pose(self.pose_paths[idx])

        idx2 = np.random.randint(len(self.pose_paths))
        rgb2 = data_io.load_rgb(self.color_paths[idx2], self.img_sidelength)
        pose2 = data_io.load_pose(self.pose_paths[idx2])

        sample = {
            "x": torch.from_numpy(rgb).float(),                 # (H,W,3)
            "x_target": torch.from_numpy(rgb2).float(),         # clean target
            "R1": torch.from_numpy(pose[:3, :3].copy()).float(),
            "R2": torch.from_numpy(pose2[:3, :3].copy()).float(),
            "t1": torch.from_numpy(pose[:3, -1].copy()).float(),
            "t2": torch.from_numpy(pose2[:3, -1].copy()).float(),
            "K": torch.from_numpy(K4[:3, :3].copy()).float(),
        }
        if self.cpu_noising:  # reference-equivalent path (data_loader.py:92-110)
            sched = _schedule()
            t = np.random.randint(0, 1000)
            noise = torch.randn_like(sample["x_target"])
            z = (sched.sqrt_alphas_cumprod[t] * sample["x_target"]
                 + sched.sqrt_one_minus_alphas_cumprod[t] * noise)
            sample["z"] = z
            sample["noise"] = noise
            sample["logsnr"] = torch.tensor(
                logsnr_schedule_cosine(t / 1000.0), dtype=torch.float32)
            sample["t"] = torch.tensor(t, dtype=torch.int64)
        return sample


class SceneClassDataset(torch.utils.data.Dataset):
    """Two-level dataset: class -> instances (reference data_loader.py:116-196).

    Keeps the reference's constructor signature and (observations,
    ground_truth) __getitem__ contract; `collate_fn` stacks every field into
    tensors (fix of D1).
    """

    def __init__(self, root_dir: str, img_sidelength: Optional[int] = None,
                 max_num_instances: int = -1,
                 max_observations_per_instance: int = -1,
                 specific_observation_idcs=None,
                 samples_per_instance: int = 2,
                 cpu_noising: bool = False):
        self.samples_per_instance = samples_per_instance
        self.instance_dirs = sorted(glob(os.path.join(root_dir, "*/")))
        if not self.instance_dirs:
            raise FileNotFoundError(
                f"no instance directories found under {root_dir!r}")
        if max_num_instances != -1:
            self.instance_dirs = self.instance_dirs[:max_num_instances]

        self.all_instances = [
            SceneInstanceDataset(instance_idx=i, instance_dir=d,
                                 specific_observation_idcs=specific_observation_idcs,
                                 img_sidelength=img_sidelength,
                                 num_images=max_observations_per_instance,
                                 cpu_noising=cpu_noising)
            for i, d in enumerate(self.instance_dirs)]
        self.num_per_instance_observations = [len(o) for o in self.all_instances]
        self.num_instances = len(self.all_instances)

    def set_img_sidelength(self, s: int) -> None:
        for inst in self.all_instances:
            inst.set_img_sidelength(s)

    def __len__(self) -> int:
        return int(np.sum(self.num_per_instance_observations))

    def get_instance_idx(self, idx: int):
        """Map a flat sample index to (instance index, observation index)
        via the cumulative observation counts."""
        bounds = np.cumsum(self.num_per_instance_observations)
        obj_idx = int(np.searchsorted(bounds, idx, side="right"))
        start = 0 if obj_idx == 0 else int(bounds[obj_idx - 1])
        return obj_idx, int(idx - start)

    def collate_fn(self, batch_list):
        """Stack list-of-(observations, ground_truth) into tensor dicts.
        Every field becomes a stacked tensor (fixes reference D1)."""
        obs_lists, gt_lists = zip(*batch_list)
        out_obs: Dict[str, torch.Tensor] = {}
        flat = [o for obs in obs_lists for o in obs]
        for k in flat[0].keys():
            out_obs[k] = torch.stack([torch.as_tensor(s[k]) for s in flat])
        flat_gt = [g for gts in gt_lists for g in gts]
        out_gt: Dict[str, torch.Tensor] = {}
        for k in flat_gt[0].keys():
            out_gt[k] = torch.stack([torch.as_tensor(s[k]) for s in flat_gt])
        return out_obs, out_gt

    def __getitem__(self, idx: int):
        obj_idx, rel_idx = self.get_instance_idx(idx)
        observations = [self.all_instances[obj_idx][rel_idx]]
        for _ in range(self.samples_per_instance - 1):
            observations.append(self.all_instances[obj_idx][
                np.random.randint(len(self.all_instances[obj_idx]))])
        ground_truth = [
            {"noise": o["noise"]} if "noise" in o else {"x_target": o["x_target"]}
            for o in observations]
        return observations, ground_truth
