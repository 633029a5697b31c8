"""Configuration layer (the reference has none — SURVEY.md §5.6).

All hyperparameters of the reference live as FLAX dataclass defaults
(/root/reference/model/xunet.py:207-215), Trainer kwargs
(/root/reference/train.py:80-89) and hardcoded literals. Here they are real
dataclasses, loadable from YAML and overridable from the CLI.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional, Sequence, Tuple


@dataclass
class XUNetConfig:
    """X-UNet architecture config.

    Defaults mirror the reference model defaults
    (/root/reference/model/xunet.py:207-215): the README "small" config.
    """

    ch: int = 32
    ch_mult: Tuple[int, ...] = (1, 2)
    emb_ch: int = 32
    num_res_blocks: int = 2
    attn_resolutions: Tuple[int, ...] = (8, 16, 32)
    attn_heads: int = 4
    dropout: float = 0.1
    use_pos_emb: bool = False
    use_ref_pose_emb: bool = False
    img_channels: int = 3

    def __post_init__(self) -> None:
        self.ch_mult = tuple(self.ch_mult)
        self.attn_resolutions = tuple(self.attn_resolutions)

    @property
    def num_resolutions(self) -> int:
        return len(self.ch_mult)

    @staticmethod
    def tiny() -> "XUNetConfig":
        """BASELINE.json config 1: CPU plumbing config."""
        return XUNetConfig(ch=8, ch_mult=(1, 2), emb_ch=8, num_res_blocks=1,
                           attn_resolutions=(8, 16), dropout=0.0)

    @staticmethod
    def small() -> "XUNetConfig":
        """BASELINE.json config 2: the README small config (64x64)."""
        return XUNetConfig()

    @staticmethod
    def full() -> "XUNetConfig":
        """BASELINE.json config 3: full 3DiM X-UNet (128x128)."""
        return XUNetConfig(ch=256, ch_mult=(1, 2, 2, 4), emb_ch=1024,
                           num_res_blocks=2, attn_resolutions=(16, 32))

    @staticmethod
    def named(name: str) -> "XUNetConfig":
        try:
            return {"tiny": XUNetConfig.tiny, "small": XUNetConfig.small,
                    "full": XUNetConfig.full}[name]()
        except KeyError:
            raise ValueError(f"unknown model config '{name}' "
                             f"(expected tiny|small|full)") from None


@dataclass
class TrainConfig:
    """Trainer config. Defaults mirror /root/reference/train.py:80-89."""

    train_batch_size: int = 2          # per-rank batch size
    train_lr: float = 1e-4
    train_num_steps: int = 100_000
    save_every: int = 1000
    img_sidelength: int = 64
    results_folder: str = "./results"
    ckpt_folder: str = "checkpoints"

    # New-framework knobs (no reference equivalent):
    loss: str = "mse"                  # "mse" | "frob" (reference train.py:67 quirk D8)
    amp: str = "bf16"                  # "bf16" | "off"
    cond_drop_prob: float = 0.1        # CFG pose-drop prob (reference train.py:64)
    log_every: int = 50
    seed: int = 0
    adam_betas: Tuple[float, float] = (0.9, 0.999)
    adam_eps: float = 1e-8
    bucket_mb: float = 40.0            # DDP gradient bucket size (xGMI sizing)
    use_graph: bool = False            # hipGraph-capture the training step
    data: str = "auto"                 # "auto" | "synthetic" | "srn"
    num_workers: int = 4
    resume: Optional[str] = None       # checkpoint path to resume from


@dataclass
class SampleConfig:
    """Sampler config. Defaults mirror /root/reference/sampling.py:55-66,128-133."""

    num_steps: int = 1000
    guidance_weight: float = 3.0
    batch_size: int = 1
    img_sidelength: int = 64
    use_hip_graph: bool = True
    clip_denoised: bool = True


def _coerce(value, ftype):
    import typing
    origin = typing.get_origin(ftype)
    if origin in (tuple, Tuple):
        return tuple(value)
    return value


def config_from_dict(cls, d: dict):
    """Build a config dataclass from a dict, ignoring unknown keys with a warning."""
    names = {f.name: f for f in dataclasses.fields(cls)}
    kwargs = {}
    for k, v in d.items():
        if k in names:
            kwargs[k] = _coerce(v, names[k].type)
        else:
            import warnings
            warnings.warn(f"{cls.__name__}: ignoring unknown config key {k!r}")
    return cls(**kwargs)


def load_yaml_config(path: str):
    """Load {model: {...}, train: {...}, sample: {...}} from a YAML file."""
    import yaml
    with open(path) as f:
        raw = yaml.safe_load(f) or {}
    return {
        "model": config_from_dict(XUNetConfig, raw.get("model", {})),
        "train": config_from_dict(TrainConfig, raw.get("train", {})),
        "sample": config_from_dict(SampleConfig, raw.get("sample", {})),
    }
