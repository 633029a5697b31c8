"""Trainer integration tests (tiny config, CPU, synthetic data)."""

import os

import pytest
import torch

from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
from novel_view_synthesis_3d_amd.engine import checkpoint as ckpt
from novel_view_synthesis_3d_amd.engine.trainer import Trainer


def make_trainer(tmp_path, steps=5, sidelength=16, lr=1e-3, **cfg_kw):
    cfg = TrainConfig()
    cfg.data = "synthetic"
    cfg.log_every = 10_000
    cfg.ckpt_folder = str(tmp_path / "ckpt")
    for k, v in cfg_kw.items():
        setattr(cfg, k, v)
    mc = XUNetConfig(ch=8, ch_mult=(1, 2), emb_ch=8, num_res_blocks=1,
                     attn_resolutions=(8,), dropout=0.0)
    return Trainer(None, train_batch_size=2, train_lr=lr,
                   train_num_steps=steps, save_every=10 ** 9,
                   img_sidelength=sidelength,
                   results_folder=str(tmp_path / "results"),
                   model_cfg=mc, train_cfg=cfg, device="cpu")


def test_loss_decreases(tmp_path):
    trainer = make_trainer(tmp_path, lr=3e-3)
    first = [float(trainer.train_step().item()) for _ in range(3)]
    for _ in range(40):
        last_loss = trainer.train_step()
    last = [float(trainer.train_step().item()) for _ in range(3)]
    assert sum(last) / 3 < sum(first) / 3, (first, last)


def test_frob_loss_matches_reference_formula(tmp_path):
    trainer = make_trainer(tmp_path, loss="frob")
    out = torch.randn(2, 8, 8, 3)
    noise = torch.randn(2, 8, 8, 3)
    got = trainer.compute_loss(out, noise)
    # reference train.py:67: mean(norm(flat)) = frobenius norm of everything
    expect = (out - noise).pow(2).sum().sqrt()
    assert torch.allclose(got, expect, atol=1e-5)


def test_checkpoint_save_resume_roundtrip(tmp_path):
    trainer = make_trainer(tmp_path)
    for _ in range(3):
        trainer.train_step()
    trainer.step = 3
    path = ckpt.save_checkpoint(trainer.cfg.ckpt_folder, trainer.model,
                                trainer.opt, trainer.step)
    assert os.path.exists(path)

    trainer2 = make_trainer(tmp_path)
    step = ckpt.load_checkpoint(path, trainer2.model, trainer2.opt)
    assert step == 3
    for p1, p2 in zip(trainer.model.parameters(), trainer2.model.parameters()):
        assert torch.equal(p1, p2)
    # Adam state restored (m/v tensors equal)
    s1 = trainer.opt.state_dict()["state"]
    s2 = trainer2.opt.state_dict()["state"]
    assert set(s1.keys()) == set(s2.keys())
    k = next(iter(s1))
    assert torch.equal(s1[k]["exp_avg"], s2[k]["exp_avg"])
    # and training continues from there
    trainer2.train_step()


def test_find_latest(tmp_path):
    trainer = make_trainer(tmp_path)
    ckpt.save_checkpoint(str(tmp_path / "ckpt"), trainer.model, trainer.opt, 1)
    p2 = ckpt.save_checkpoint(str(tmp_path / "ckpt"), trainer.model,
                              trainer.opt, 2)
    assert ckpt.find_latest(str(tmp_path / "ckpt")) == p2


def test_cond_mask_fresh_per_step(tmp_path):
    """Reference defect D2: the CFG mask was frozen at trace time. Ours must
    differ across steps."""
    trainer = make_trainer(tmp_path, cond_drop_prob=0.5)
    masks = []
    for _ in range(6):
        raw = trainer.next_batch()
        _, cond_mask, _ = trainer.prepare_model_inputs(raw)
        masks.append(cond_mask)
    stacked = torch.stack(masks)
    assert stacked.std() > 0 or not torch.all(stacked == stacked[0])
