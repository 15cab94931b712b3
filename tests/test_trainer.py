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


def test_grad_accumulation_equivalence(tmp_path):
    """Two micro-batches at grad_accumulation=2 produce the same params
    as one optimizer step on the concatenated batch (the unsup loss is
    numValidPixels-normalized, so the mean of equal halves equals the
    whole)."""
    import torch

    from deepof_amd.config import Config
    from deepof_amd.data import SyntheticFlowDataset
    from deepof_amd.engine import Trainer

    def make(accum, run):
        cfg = Config.from_dict(dict(
            dataset="synthetic", image_size=(48, 64), batch_size=2,
            num_workers=0, model="flownets", precision="fp32",
            device="cpu", log_dir=str(tmp_path), run_name=run,
            grad_accumulation=accum, resume=False, seed=0,
        ))
        return Trainer(cfg)

    ds = SyntheticFlowDataset(4, 48, 64)
    b1 = {k: torch.stack([ds[0][k], ds[1][k]]) for k in ("img1", "img2")}
    b2 = {k: torch.stack([ds[2][k], ds[3][k]]) for k in ("img1", "img2")}
    both = {k: torch.cat([b1[k], b2[k]]) for k in b1}

    t_acc = make(2, "acc")
    t_one = make(1, "one")
    # identical init (same seed) — sanity
    for pa, pb in zip(t_acc.model.parameters(), t_one.model.parameters()):
        assert torch.equal(pa, pb)

    t_acc.train_step(b1)
    t_acc.train_step(b2)          # boundary: one optimizer step
    t_one.train_step(both)        # one step on the full batch

    # fp summation order differs between half-batch means and the whole
    # batch; Adam's rsqrt amplifies that on near-zero second moments
    for pa, pb in zip(t_acc.model.parameters(), t_one.model.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-3, atol=1e-4)


def test_best_checkpoint_saved(tmp_path):
    """ckpt_best.pt tracks the lowest eval AEE (model-only, with the
    achieving epoch recorded)."""
    import os

    import torch

    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(48, 64), batch_size=2,
        num_workers=0, model="flownets", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="best", eval_interval_epochs=1,
        max_epochs=1, resume=False,
    ))
    tr = Trainer(cfg)
    # shrink the epoch: 4 samples -> 2 steps
    from deepof_amd.data import SyntheticFlowDataset

    import deepof_amd.engine.trainer as T

    orig = T.build_datasets

    def tiny(cfg_):
        return (SyntheticFlowDataset(4, 48, 64),
                SyntheticFlowDataset(4, 48, 64, seed=1))

    T.build_datasets = tiny
    try:
        tr.fit(max_epochs=1)
    finally:
        T.build_datasets = orig
    best = os.path.join(str(tmp_path), "best", "ckpt_best.pt")
    assert os.path.exists(best)
    state = torch.load(best, weights_only=False)
    assert state["aee"] == tr.best_aee
    assert "model" in state and state["epoch"] == 1


import pytest


@pytest.mark.parametrize("model", ["flownets", "flownetc", "vgg16",
                                   "inception_v3"])
def test_every_model_family_trains(tmp_path, model):
    """One real optimizer step through Trainer for every encoder family
    (decoder wiring, loss pyramid, Adam) on CPU."""
    import numpy as np

    from deepof_amd.config import Config
    from deepof_amd.data import SyntheticFlowDataset
    from deepof_amd.engine import Trainer

    import torch

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(64, 96), batch_size=1,
        num_workers=0, model=model, precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name=f"fam_{model}", resume=False,
    ))
    tr = Trainer(cfg)
    ds = SyntheticFlowDataset(2, 64, 96)
    batch = {k: v.unsqueeze(0) for k, v in ds[0].items()}
    before = [p.detach().clone() for p in tr.model.parameters()]
    parts = tr.train_step(batch)
    assert np.isfinite(parts["total"])
    changed = any(not torch.equal(a, b) for a, b in
                  zip(before, tr.model.parameters()))
    assert changed, f"{model}: no parameter moved after a step"


def test_per_scale_loss_logging(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.data import SyntheticFlowDataset
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(48, 64), batch_size=2,
        num_workers=0, model="flownets", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="ps", resume=False,
    ))
    tr = Trainer(cfg)
    ds = SyntheticFlowDataset(2, 48, 64)
    batch = {k: v.unsqueeze(0) for k, v in ds[0].items()}
    parts = tr.train_step(batch, log_scales=True)
    assert len(parts["scale_losses"]) == 6  # one per pyramid scale
    assert all(s >= 0 for s in parts["scale_losses"])
    parts2 = tr.train_step(batch)
    assert "scale_losses" not in parts2


def test_cpu_training_loss_decreases(tmp_path):
    """30 optimizer steps on one repeated synthetic batch must reduce
    the unsupervised loss (CPU counterpart of the GPU convergence
    smoke) — catches sign/scale regressions in the loss/optimizer
    wiring without a GPU."""
    import numpy as np

    from deepof_amd.config import Config
    from deepof_amd.data import SyntheticFlowDataset
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(32, 48), batch_size=2,
        num_workers=0, model="flownets", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="conv", lr=1e-4, resume=False,
    ))
    tr = Trainer(cfg)
    ds = SyntheticFlowDataset(2, 32, 48)
    batch = {k: torch.stack([ds[0][k], ds[1][k]]) for k in ds[0]}
    losses = [tr.train_step(batch)["total"] for _ in range(30)]
    assert all(np.isfinite(l) for l in losses)
    assert min(losses[-5:]) < losses[0], (losses[0], losses[-5:])

