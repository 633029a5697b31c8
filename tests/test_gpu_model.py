"""GPU integration tests: model forward/backward, trainer step, sampler graph.

These run the REAL MI355X path: HIP kernels active (no eager fallback), bf16
autocast, FusedAdam.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _mk_trainer(tmp_path, model="small", H=64, batch=2):
    from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
    from novel_view_synthesis_3d_amd.engine.trainer import Trainer
    cfg = TrainConfig()
    cfg.data = "synthetic"
    cfg.log_every = 10 ** 9
    cfg.ckpt_folder = str(tmp_path / "ckpt")
    return Trainer(None, train_batch_size=batch, train_lr=1e-3,
                   train_num_steps=10 ** 9, img_sidelength=H,
                   results_folder=str(tmp_path / "res"),
                   model_cfg=XUNetConfig.named(model), train_cfg=cfg)


def test_small_model_train_step_and_loss_decreases(tmp_path):
    trainer = _mk_trainer(tmp_path)
    first = [float(trainer.train_step().item()) for _ in range(3)]
    for _ in range(30):
        trainer.train_step()
    last = [float(trainer.train_step().item()) for _ in range(3)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(first + last)))
    assert sum(last) < sum(first), (first, last)


def test_model_gpu_matches_cpu_eager_fp32(tmp_path):
    """End-to-end numerics: full fp32 forward on GPU (HIP kernels) vs CPU
    (eager oracle) on identical weights + inputs."""
    from novel_view_synthesis_3d_amd.config import XUNetConfig
    from novel_view_synthesis_3d_amd.models.xunet import XUNet
    from test_model import make_inputs  # tests/ is on sys.path via conftest

    torch.manual_seed(0)
    cfg = XUNetConfig(ch=32, ch_mult=(1, 2), emb_ch=32, num_res_blocks=1,
                      attn_resolutions=(16,), dropout=0.0)
    model = XUNet(cfg, img_sidelength=32)
    with torch.no_grad():  # nonzero head so outputs are informative
        model.Conv_1.weight.normal_(0, 0.05)
    model.eval()
    batch, _ = make_inputs(B=2, H=32)
    out_cpu = model(batch, cond_mask=torch.ones(2))

    model_gpu = model.cuda()
    batch_gpu = {k: v.cuda() for k, v in batch.items()}
    out_gpu = model_gpu(batch_gpu, cond_mask=torch.ones(2, device="cuda"))
    err = (out_gpu.cpu() - out_cpu).abs().max().item()
    scale = out_cpu.abs().max().item()
    assert err < 5e-3 * max(scale, 1.0), (err, scale)


def test_sampler_graph_capture_matches_eager(tmp_path):
    from novel_view_synthesis_3d_amd.config import XUNetConfig
    from novel_view_synthesis_3d_amd.data.synthetic import synthetic_batch
    from novel_view_synthesis_3d_amd.diffusion.sampler import DDPMSampler
    from novel_view_synthesis_3d_amd.models.xunet import XUNet

    torch.manual_seed(0)
    cfg = XUNetConfig(ch=32, ch_mult=(1, 2), emb_ch=32, num_res_blocks=1,
                      attn_resolutions=(16,), dropout=0.0)
    model = XUNet(cfg, img_sidelength=32).cuda()
    with torch.no_grad():
        model.Conv_1.weight.normal_(0, 0.05)
    g = torch.Generator(device="cuda").manual_seed(0)
    cond = synthetic_batch(2, 32, "cuda", g)
    cond.pop("x_target")
    z0 = torch.randn(2, 32, 32, 3, device="cuda", generator=g)

    out_eager = DDPMSampler(model, num_steps=6, use_graph=False).sample(
        cond, z_init=z0)
    out_graph = DDPMSampler(model, num_steps=6, use_graph=True).sample(
        cond, z_init=z0)
    assert torch.isfinite(out_graph).all()
    # (graph replays advance the philox offset, so outputs are not
    # reseedable via manual_seed — check consistency, not equality)
    assert out_eager.shape == out_graph.shape
    assert out_graph.abs().max().item() < 50.0
    # distributions should agree: same model, same schedule
    assert abs(out_graph.std().item() - out_eager.std().item()) < 0.5


def test_checkpoint_roundtrip_gpu(tmp_path):
    from novel_view_synthesis_3d_amd.engine import checkpoint as ckpt
    trainer = _mk_trainer(tmp_path)
    for _ in range(2):
        trainer.train_step()
    path = ckpt.save_checkpoint(str(tmp_path / "ckpt"), trainer.model,
                                trainer.opt, 2)
    trainer2 = _mk_trainer(tmp_path)
    step = ckpt.load_checkpoint(path, trainer2.model, trainer2.opt,
                                map_location="cuda")
    assert step == 2
    trainer2.train_step()
    torch.cuda.synchronize()


def test_trainer_graph_mode(tmp_path):
    """hipGraph-captured training step: captures, replays, finite losses."""
    from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
    from novel_view_synthesis_3d_amd.engine.trainer import Trainer
    cfg = TrainConfig()
    cfg.data = "synthetic"
    cfg.log_every = 10 ** 9
    cfg.use_graph = True
    cfg.ckpt_folder = str(tmp_path / "ckpt")
    trainer = Trainer(None, train_batch_size=2, train_lr=1e-3,
                      train_num_steps=10 ** 9, img_sidelength=64,
                      results_folder=str(tmp_path / "res"),
                      model_cfg=XUNetConfig.named("small"), train_cfg=cfg)
    losses = [float(trainer.train_step().item()) for _ in range(4)]
    torch.cuda.synchronize()
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert getattr(trainer, "graph_active", False), \
        "graph capture silently fell back"
    # replays produce varying losses (fresh philox noise per replay)
    assert len(set(losses)) > 1, losses


def test_trainer_graph_mode_full_config(tmp_path):
    """FULL-config graph training: the round-1 claim failed exactly here
    (replay on a stream other than the capture stream corrupted the loss
    buffer when replays queue without host syncs). Queue several replays
    back-to-back bench.py-style, sync once, and require a finite,
    plausible (non-negative MSE) loss."""
    from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
    from novel_view_synthesis_3d_amd.engine.trainer import Trainer
    cfg = TrainConfig()
    cfg.data = "synthetic"
    cfg.log_every = 10 ** 9
    cfg.use_graph = True
    cfg.seed = 1234
    cfg.ckpt_folder = str(tmp_path / "ckpt")
    trainer = Trainer(None, train_batch_size=4,
                      train_num_steps=10 ** 9, img_sidelength=128,
                      results_folder=str(tmp_path / "res"),
                      model_cfg=XUNetConfig.named("full"), train_cfg=cfg)
    for _ in range(3):
        trainer.train_step()
    torch.cuda.synchronize()
    for _ in range(5):  # no host sync between replays
        loss = trainer.train_step()
    torch.cuda.synchronize()
    lv = float(loss.item())
    assert getattr(trainer, "graph_active", False)
    assert lv == lv and 0.0 <= lv < 10.0, lv  # finite, valid MSE
    wn = float(trainer.model.Conv_0.weight.norm().item())
    assert wn == wn, "params corrupted"
