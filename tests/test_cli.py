"""CLI end-to-end on CPU: train -> eval -> infer round trip."""

import os
import subprocess
import sys

import numpy as np
from PIL import Image


def _run(args, cwd):
    return subprocess.run([sys.executable, "-m", "deepof_amd"] + args,
                          capture_output=True, text=True, cwd=cwd)


def test_cli_train_eval_infer(tmp_path):
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = _run(["train", "--max-steps", "2",
              "dataset=synthetic", "image_size=[48,64]", "batch_size=2",
              "num_workers=0", "precision=fp32", "device=cpu",
              f"log_dir={tmp_path}", "run_name=cli", "log_interval=1"],
             repo)
    assert r.returncode == 0, r.stderr[-2000:]
    ckpt = os.path.join(str(tmp_path), "cli", "ckpt_last.pt")
    assert os.path.exists(ckpt)

    rng = np.random.default_rng(0)
    for name in ("a.png", "b.png"):
        Image.fromarray(rng.integers(0, 255, (48, 64, 3),
                                     dtype=np.uint8)).save(tmp_path / name)
    out = str(tmp_path / "flow")
    r = _run(["infer", "--checkpoint", ckpt,
              "--img1", str(tmp_path / "a.png"),
              "--img2", str(tmp_path / "b.png"),
              "--out", out, "device=cpu"], repo)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out + ".flo") and os.path.exists(out + ".jpg")

    r = _run(["eval", "--checkpoint", ckpt,
              "dataset=synthetic", "image_size=[48,64]", "batch_size=2",
              "num_workers=0", "device=cpu", f"log_dir={tmp_path}"], repo)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "AEE:" in r.stdout


def test_bench_contract_single_process():
    """bench.py (the driver contract) runs on CPU and prints ONE valid
    JSON line with the required fields."""
    import json

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch", "1", "--height", "64", "--width", "64",
         "--dtype", "fp32", "--no-channels-last", "--no-graphs"],
        capture_output=True, text=True, cwd=repo, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in rec, field
    assert rec["n_gpus"] == 1 and rec["scaling"] == "weak"
    assert rec["config"]["parallelism"] == "dp1"
