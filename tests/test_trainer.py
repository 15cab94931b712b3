import os

import torch

from deepof_amd.config import Config
from deepof_amd.engine import Trainer


def _cfg(tmp_path, **kw):
    base = dict(
        dataset="synthetic",
        image_size=(64, 96),
        batch_size=2,
        num_workers=0,
        model="flownets",
        precision="fp32",
        device="cpu",
        log_dir=str(tmp_path),
        run_name="t",
        log_interval=1,
        save_interval_epochs=1,
    )
    base.update(kw)
    return Config.from_dict(base)


def test_trainer_steps_and_checkpoint(tmp_path):
    cfg = _cfg(tmp_path)
    tr = Trainer(cfg)
    tr.fit(max_steps=2)
    assert tr.global_step == 2
    ckpt = os.path.join(str(tmp_path), "t", "ckpt_last.pt")
    assert os.path.exists(ckpt)
    state = torch.load(ckpt, weights_only=False)
    assert state["global_step"] == 2

    # resume picks up where we left off
    tr2 = Trainer(cfg)
    assert tr2.global_step == 2


def test_trainer_loss_decreases(tmp_path):
    cfg = _cfg(tmp_path, lr=1e-4, run_name="t2")
    tr = Trainer(cfg)
    train_ds, _ = __import__(
        "deepof_amd.engine.trainer", fromlist=["build_datasets"]
    ).build_datasets(cfg)
    batch = {k: v.unsqueeze(0).repeat(2, *[1] * v.dim())
             for k, v in train_ds[0].items()}
    losses = [tr.train_step(batch)["total"] for _ in range(6)]
    assert losses[-1] < losses[0]


def test_guided_trainer_step(tmp_path):
    cfg = _cfg(tmp_path, guided=True, run_name="t3")
    tr = Trainer(cfg)
    train_ds, _ = __import__(
        "deepof_amd.engine.trainer", fromlist=["build_datasets"]
    ).build_datasets(cfg)
    batch = {k: v.unsqueeze(0) for k, v in train_ds[0].items()}
    parts = tr.train_step(batch)
    assert "guided" in parts and "unsup" in parts


def test_lr_schedule(tmp_path):
    cfg = _cfg(tmp_path, lr=1.6e-5, epochs_per_decay=18, run_name="t4")
    tr = Trainer(cfg)
    assert tr.current_lr() == 1.6e-5
    tr.epoch = 18
    assert abs(tr.current_lr() - 0.8e-5) < 1e-12
    tr.epoch = 36
    assert abs(tr.current_lr() - 0.4e-5) < 1e-12
