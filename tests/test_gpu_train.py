"""GPU end-to-end engine validation (trainer, eval, FlowNetC)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_trainer_gpu_synthetic_loss_decreases(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(192, 256), batch_size=8,
        num_workers=2, model="flownets", precision="bf16", device="cuda",
        channels_last=True, log_dir=str(tmp_path), run_name="g",
        lr=2e-4, log_interval=10,
    ))
    tr = Trainer(cfg)
    from deepof_amd.engine.trainer import build_datasets
    train_ds, _ = build_datasets(cfg)
    batch = {k: torch.stack([train_ds[i][k] for i in range(8)])
             for k in train_ds[0]}
    losses = [tr.train_step(batch)["total"] for _ in range(30)]
    assert losses[-1] == losses[-1]  # finite
    assert min(losses[-5:]) < losses[0], losses[:3] + losses[-3:]


def test_trainer_gpu_guided(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(128, 192), batch_size=4,
        num_workers=0, model="flownets", precision="bf16", device="cuda",
        channels_last=True, guided=True, log_dir=str(tmp_path),
        run_name="gg", log_interval=10,
    ))
    tr = Trainer(cfg)
    from deepof_amd.engine.trainer import build_datasets
    train_ds, _ = build_datasets(cfg)
    batch = {k: torch.stack([train_ds[i][k] for i in range(4)])
             for k in train_ds[0]}
    parts = tr.train_step(batch)
    assert "guided" in parts and parts["total"] == parts["total"]
