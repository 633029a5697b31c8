"""bench.py driver-contract tests: single-process and torchrun world=2 (gloo).

The round driver launches `python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...` — this must
produce exactly ONE JSON line from rank 0 with the contract fields.
"""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _parse_json_line(stdout: str) -> dict:
    lines = [l for l in stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {stdout!r}"
    return json.loads(lines[0])


REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


@pytest.mark.timeout(600)
def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--sidelength", "16",
         "--batch", "2", "--steps", "2", "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=540)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _parse_json_line(out.stdout)
    for k in REQUIRED:
        assert k in rec, k
    assert rec["n_gpus"] == 1
    assert rec["value"] > 0
    assert rec["scaling"] == "weak"
    assert rec["config"]["global_batch"] == 2


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_contract():
    env = dict(os.environ)
    env.pop("RANK", None); env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--model", "tiny", "--sidelength", "16", "--batch", "2",
         "--steps", "2", "--warmup", "1"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=840)
    assert out.returncode == 0, out.stderr[-3000:]
    rec = _parse_json_line(out.stdout)
    assert rec["n_gpus"] == 2
    assert rec["config"]["global_batch"] == 4  # 2 per rank x 2 ranks
    assert rec["config"]["parallelism"] == "dp2"
