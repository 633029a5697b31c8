"""CLI surface tests: train.py and sample.py run end-to-end (tiny, CPU)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_train_cli_tiny_synthetic(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "train.py"),
         "--folder", str(tmp_path / "nodata"),
         "--model", "tiny", "--sidelength", "16", "--batch-size", "2",
         "--num-steps", "3", "--save-every", "2", "--data", "synthetic",
         "--results-folder", str(tmp_path / "results")],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=540)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "training completed" in out.stdout
    # jsonl log written
    log = tmp_path / "results" / "train_log.jsonl"
    assert log.exists()
    rec = json.loads(log.read_text().splitlines()[0])
    assert "loss" in rec and "images_per_sec" in rec


@pytest.mark.timeout(600)
def test_train_then_sample_cli(tmp_path):
    ckpt_dir = tmp_path / "ckpt"
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-c", f"""
import sys; sys.path.insert(0, {str(ROOT)!r})
import os; os.chdir({str(tmp_path)!r})
from novel_view_synthesis_3d_amd.config import TrainConfig, XUNetConfig
from novel_view_synthesis_3d_amd.engine.trainer import Trainer
cfg = TrainConfig(); cfg.data = 'synthetic'; cfg.ckpt_folder = {str(ckpt_dir)!r}
mc = XUNetConfig.named('tiny')  # must match sample.py --model tiny
t = Trainer(None, train_batch_size=2, train_num_steps=2, save_every=10,
            img_sidelength=16, results_folder={str(tmp_path / 'res')!r},
            model_cfg=mc, train_cfg=cfg, device='cpu')
t.train()
"""], capture_output=True, text=True, timeout=240, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    ckpts = list(ckpt_dir.glob("model_*.pt"))
    assert ckpts, "no checkpoint written"

    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "sample.py"),
         "--checkpoint", str(sorted(ckpts)[-1]),
         "--model", "tiny", "--sidelength", "16", "--steps", "3",
         "--batch-size", "1", "--out", str(tmp_path / "samples"),
         "--no-graph"],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=300,
        env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    outs = (list((tmp_path / "samples").glob("*.png"))
            + list((tmp_path / "samples").glob("*.npy")))
    assert outs, "no sample written"
