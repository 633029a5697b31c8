"""The pose-conditional X-UNet (3DiM, arXiv:2210.04628, k=1 conditioning).

Re-designed from /root/reference/model/xunet.py:142-280 for MI355X:
activations are (B, F=2, H, W, C) contiguous (NHWC + frame axis), all hot ops
dispatch to CDNA4 HIP kernels through novel_view_synthesis_3d_amd.ops.

batch schema (reference train.py:53-60):
  x      (B, H, W, 3)  clean source view in [-1, 1]
  z      (B, H, W, 3)  noised target view
  logsnr (B,)          logsnr of z's noise level
  R1, R2 (B, 3, 3)     cam->world rotations (source, target)
  t1, t2 (B, 3)        camera origins (world)
  K      (B, 3, 3)     shared pinhole intrinsics
forward returns predicted noise for the TARGET frame only: (B, H, W, 3)
(reference xunet.py:280 `return conv(h)[:, 1]`).

Unlike the reference (defect D7 fixed) GroupNorm uses min(32, C) groups, so
tiny smoke configs run; everything else matches the reference math including
the 1/sqrt(2) residual scales, zero-init output convs, missing attention
output projection, and joint-frame GroupNorm statistics.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F
from torch import nn

from novel_view_synthesis_3d_amd import ops
from novel_view_synthesis_3d_amd.config import XUNetConfig
from novel_view_synthesis_3d_amd.models.layers import (
    AttnBlock, Dense, FrameConv, JointGroupNorm, ResnetBlock, XUNetBlock,
)
from novel_view_synthesis_3d_amd.utils.init import normal_

POSE_EMB_DIM = 3 * (1 + 2 * 15) + 3 * (1 + 2 * 8)  # 93 + 51 = 144


class ConditioningProcessor(nn.Module):
    """logsnr + camera-pose conditioning (/root/reference/model/xunet.py:142-203).

    * logsnr -> squash -> DDPM posenc -> Dense_0 -> silu -> Dense_1
    * poses -> per-pixel rays -> NeRF posenc (pos deg 15, dir deg 8) -> 144ch
    * classifier-free-guidance masking of the pose embedding
    * optional learned pos_emb (H,W,144) and per-frame ref_pose_emb (144,)
    * per-resolution strided FrameConv 144 -> emb_ch
    """

    flax_type = "ConditioningProcessor"

    def __init__(self, cfg: XUNetConfig, img_sidelength: int):
        super().__init__()
        self.cfg = cfg
        self.H = self.W = img_sidelength
        D = POSE_EMB_DIM
        self.Dense_0 = Dense(cfg.emb_ch, cfg.emb_ch)
        self.Dense_1 = Dense(cfg.emb_ch, cfg.emb_ch)
        if cfg.use_pos_emb:
            self.pos_emb = nn.Parameter(
                normal_(torch.empty(self.H, self.W, D), std=1.0 / math.sqrt(D)))
        if cfg.use_ref_pose_emb:
            self.ref_pose_emb_first = nn.Parameter(
                normal_(torch.empty(D), std=1.0 / math.sqrt(D)))
            self.ref_pose_emb_other = nn.Parameter(
                normal_(torch.empty(D), std=1.0 / math.sqrt(D)))
        for i in range(cfg.num_resolutions):
            self.add_module(f"Conv_{i}", FrameConv(D, cfg.emb_ch, stride=2 ** i))

    def pose_features(self, batch: Dict[str, torch.Tensor],
                      cond_mask: torch.Tensor):
        """Step-invariant conditioning: rays -> NeRF posenc -> CFG mask ->
        per-level strided convs (K13+K14+K15+K2, one fused kernel + convs).
        The sampler computes this ONCE per image and replays only the
        logsnr path per DDPM step (the reference recomputes everything in
        all 2000 forwards, sampling.py:128-134)."""
        cfg = self.cfg
        B, H, W, _ = batch["x"].shape
        assert cond_mask.shape == (B,), cond_mask.shape
        emb_dtype = batch["x"].dtype
        if batch["x"].is_cuda and torch.is_autocast_enabled():
            emb_dtype = torch.get_autocast_dtype("cuda")
        R2 = torch.stack([batch["R1"], batch["R2"]], dim=1)
        t2 = torch.stack([batch["t1"], batch["t2"]], dim=1)
        pose_emb = ops.pose_embedding(R2, t2, batch["K"], cond_mask,
                                      H, W, emb_dtype)  # (B,2,H,W,144)
        if cfg.use_pos_emb:
            pose_emb = pose_emb + self.pos_emb[None, None]
        if cfg.use_ref_pose_emb:
            pose_emb = pose_emb + torch.stack(
                [self.ref_pose_emb_first, self.ref_pose_emb_other]
            ).reshape(1, 2, 1, 1, -1)

        return [getattr(self, f"Conv_{i}")(pose_emb)
                for i in range(cfg.num_resolutions)]

    def forward(self, batch: Dict[str, torch.Tensor], cond_mask: torch.Tensor,
                pose_cache=None):
        cfg = self.cfg
        # logsnr embedding (K12 + K6)
        logsnr = ops.squash_logsnr(batch["logsnr"])
        logsnr_emb = ops.posenc_ddpm(logsnr, emb_ch=cfg.emb_ch, max_time=1.0)
        logsnr_emb = logsnr_emb.to(batch["x"].dtype)
        logsnr_emb = self.Dense_1(F.silu(self.Dense_0(logsnr_emb)))

        pose_embs = (pose_cache if pose_cache is not None
                     else self.pose_features(batch, cond_mask))
        return logsnr_emb, pose_embs


class XUNet(nn.Module):
    """X-UNet (/root/reference/model/xunet.py:205-280).

    Because torch modules are constructed with explicit shapes (flax infers
    them from a traced sample), the constructor takes `img_sidelength` — it
    determines which blocks get attention (resolution in attn_resolutions),
    exactly as the reference's trace-time `h.shape[2]` checks do.
    """

    flax_type = "XUNet"

    def __init__(self, cfg: Optional[XUNetConfig] = None, img_sidelength: int = 64):
        super().__init__()
        cfg = cfg or XUNetConfig()
        self.cfg = cfg
        self.img_sidelength = img_sidelength
        L = cfg.num_resolutions
        nrb = cfg.num_res_blocks

        self.ConditioningProcessor_0 = ConditioningProcessor(cfg, img_sidelength)
        self.Conv_0 = FrameConv(cfg.img_channels, cfg.ch)  # stem

        def res_at(level: int) -> int:
            return img_sidelength // (2 ** level)

        # Down path — mirror the reference's construction loop and skip-stack
        # channel bookkeeping (xunet.py:231-246).
        xb, rb = 0, 0  # per-type counters at XUNet scope
        self.down_names: List[List[str]] = []
        hs_ch: List[int] = [cfg.ch]
        ch_cur = cfg.ch
        for i_level in range(L):
            feats = cfg.ch * cfg.ch_mult[i_level]
            names = []
            for _ in range(nrb):
                use_attn = res_at(i_level) in cfg.attn_resolutions
                name = f"XUNetBlock_{xb}"; xb += 1
                self.add_module(name, XUNetBlock(
                    ch_cur, cfg.emb_ch, feats, cfg.attn_heads, cfg.dropout,
                    use_attn))
                names.append(name)
                ch_cur = feats
                hs_ch.append(ch_cur)
            self.down_names.append(names)
            if i_level != L - 1:
                name = f"ResnetBlock_{rb}"; rb += 1
                self.add_module(name, ResnetBlock(
                    ch_cur, cfg.emb_ch, dropout=cfg.dropout, resample="down"))
                self.down_names[-1].append(name)
                hs_ch.append(ch_cur)

        # Middle (xunet.py:248-255)
        use_attn = res_at(L - 1) in cfg.attn_resolutions
        self.mid_name = f"XUNetBlock_{xb}"; xb += 1
        self.add_module(self.mid_name, XUNetBlock(
            ch_cur, cfg.emb_ch, cfg.ch * cfg.ch_mult[L - 1], cfg.attn_heads,
            cfg.dropout, use_attn))
        ch_cur = cfg.ch * cfg.ch_mult[L - 1]

        # Up path (xunet.py:256-271)
        self.up_names: List[List[str]] = []
        for i_level in reversed(range(L)):
            feats = cfg.ch * cfg.ch_mult[i_level]
            names = []
            for _ in range(nrb + 1):
                skip_ch = hs_ch.pop()
                use_attn = res_at(i_level) in cfg.attn_resolutions
                name = f"XUNetBlock_{xb}"; xb += 1
                self.add_module(name, XUNetBlock(
                    ch_cur + skip_ch, cfg.emb_ch, feats, cfg.attn_heads,
                    cfg.dropout, use_attn))
                names.append(name)
                ch_cur = feats
            if i_level != 0:
                name = f"ResnetBlock_{rb}"; rb += 1
                self.add_module(name, ResnetBlock(
                    ch_cur, cfg.emb_ch, dropout=cfg.dropout, resample="up"))
                names.append(name)
            self.up_names.append(names)
        assert not hs_ch

        # End (xunet.py:274-280)
        self.GroupNorm_0 = JointGroupNorm(ch_cur)
        self.Conv_1 = FrameConv(ch_cur, cfg.img_channels, zero_init=True)

    # -----------------------------------------------------------------
    def forward(self, batch: Dict[str, torch.Tensor],
                cond_mask: torch.Tensor, pose_cache=None) -> torch.Tensor:
        cfg = self.cfg
        L = cfg.num_resolutions
        nrb = cfg.num_res_blocks
        logsnr_emb, pose_embs = self.ConditioningProcessor_0(
            batch, cond_mask, pose_cache=pose_cache)

        _emb_cache: Dict[int, torch.Tensor] = {}

        def emb_at(level: int) -> torch.Tensor:
            # silu((B,1,1,1,E) + (B,2,H',W',E))  (xunet.py:233,59) — the
            # silu belongs to FiLM's input; cached per level since every
            # block of a level consumes the identical tensor.
            if level not in _emb_cache:
                e = logsnr_emb[:, None, None, None, :] + pose_embs[level]
                _emb_cache[level] = F.silu(e)
            return _emb_cache[level]

        h = torch.stack([batch["x"], batch["z"]], dim=1)  # (B,2,H,W,3)
        h = self.Conv_0(h)
        hs = [h]
        for i_level in range(L):
            names = self.down_names[i_level]
            for name in names[:nrb]:
                h = getattr(self, name)(h, emb_at(i_level))
                hs.append(h)
            if len(names) > nrb:  # down-resample block
                h = getattr(self, names[nrb])(h, emb_at(i_level + 1))
                hs.append(h)

        h = getattr(self, self.mid_name)(h, emb_at(L - 1))

        for idx, i_level in enumerate(reversed(range(L))):
            names = self.up_names[idx]
            for name in names[:nrb + 1]:
                h = torch.cat([h, hs.pop()], dim=-1)
                h = getattr(self, name)(h, emb_at(i_level))
            if len(names) > nrb + 1:  # up-resample block
                h = getattr(self, names[nrb + 1])(h, emb_at(i_level - 1))
        assert not hs

        h = self.GroupNorm_0(h, silu=True)
        h = self.Conv_1(h)
        return h[:, 1]

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())
