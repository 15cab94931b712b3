"""Capability-parity tests: UCF101 joint models, Sintel multi-frame
loss, augmentation, VGG warm start, eval artifact dumps, NaN restart."""

import numpy as np
import pytest
import torch

from deepof_amd.losses import MultiFrameUnsupLoss
from deepof_amd.models import STBaseline, STSingle, build_model


def test_st_single_joint_outputs():
    m = STSingle(input_hw=(64, 96), num_classes=11)
    x = torch.randn(2, 6, 64, 96)
    flows, logits = m(x)
    assert len(flows) == 5
    assert logits.shape == (2, 11)
    loss = sum(f.abs().mean() for f in flows) + logits.pow(2).mean()
    loss.backward()


def test_st_baseline_joint_outputs():
    m = STBaseline(input_hw=(64, 64), num_classes=7)
    x = torch.randn(1, 6, 64, 64)
    flows, logits = m(x)
    assert len(flows) == 6
    assert logits.shape == (1, 7)


def test_multiframe_loss():
    torch.manual_seed(0)
    T = 4
    model, scales, weights = build_model("inception_v3", time_step=T)
    vol = torch.rand(1, 3 * T, 64, 96) * 255
    mean = torch.tensor([70.1433, 83.1915, 92.8827]).repeat(T).view(1, -1, 1, 1)
    x = (vol - mean) / 255.0
    flows = model(x)
    loss_fn = MultiFrameUnsupLoss(scales, weights)
    res = loss_fn(flows, vol)
    assert torch.isfinite(res["total"])
    res["total"].backward()
    gsum = sum(p.grad.abs().sum() for p in model.parameters()
               if p.grad is not None)
    assert torch.isfinite(gsum)


def test_augment_pair_shapes_and_ranges():
    from deepof_amd.utils.augment import augment_pair

    torch.manual_seed(0)
    img1 = torch.rand(4, 3, 32, 48) * 255
    img2 = torch.rand(4, 3, 32, 48) * 255
    geo1, geo2, ph1, ph2 = augment_pair(img1, img2)
    for t in (geo1, geo2, ph1, ph2):
        assert t.shape == img1.shape
        assert torch.isfinite(t).all()
    assert 0 <= ph1.min() and ph1.max() <= 255
    # geo transform is shared within a pair: warping both frames by the
    # same grid preserves their relative displacement statistics
    assert not torch.equal(geo1, img1)


def test_geometric_flip_only_is_exact():
    from deepof_amd.utils.augment import geometric_augment

    torch.manual_seed(3)
    img = torch.rand(2, 3, 16, 16) * 255
    g1, g2 = geometric_augment(img, img, translate=0.0,
                               scale_range=(1.0, 1.0), flip_prob=0.0)
    torch.testing.assert_close(g1, img, rtol=1e-4, atol=1e-3)


def test_vgg_warmstart(tmp_path):
    from deepof_amd.models.vgg16 import VGG16Encoder
    from deepof_amd.utils.warmstart import load_vgg16_npz

    rng = np.random.default_rng(0)
    spec = {
        "conv1_1": (3, 64), "conv1_2": (64, 64),
        "conv2_1": (64, 128), "conv2_2": (128, 128),
        "conv3_1": (128, 256), "conv3_2": (256, 256), "conv3_3": (256, 256),
        "conv4_1": (256, 512), "conv4_2": (512, 512), "conv4_3": (512, 512),
        "conv5_1": (512, 512), "conv5_2": (512, 512), "conv5_3": (512, 512),
    }
    arrs = {}
    for name, (cin, cout) in spec.items():
        arrs[f"{name}_W"] = rng.standard_normal((3, 3, cin, cout)).astype(np.float32)
        arrs[f"{name}_b"] = rng.standard_normal(cout).astype(np.float32)
    path = tmp_path / "vgg16_weights.npz"
    np.savez(path, **arrs)

    enc = VGG16Encoder(in_channels=6)
    n = load_vgg16_npz(enc, str(path))
    assert n == 26
    # conv1_1 duplicated across the 6 input channels, halved
    w = [m for m in enc.block1.modules()
         if isinstance(m, torch.nn.Conv2d)][0].weight
    expect = torch.from_numpy(arrs["conv1_1_W"]).permute(3, 2, 0, 1) * 0.5
    torch.testing.assert_close(w[:, :3], expect)
    torch.testing.assert_close(w[:, 3:], expect)


def test_eval_dump_artifacts(tmp_path):
    import os

    from deepof_amd.data import SyntheticFlowDataset, build_dataloader
    from deepof_amd.engine.evaluator import evaluate_aee
    from deepof_amd.models import build_model

    model, scales, _ = build_model("flownets")
    ds = SyntheticFlowDataset(2, 64, 64)
    dl = build_dataloader(ds, 1, shuffle=False, num_workers=0,
                          drop_last=False)
    aee = evaluate_aee(model, dl, ds.mean_bgr, scales[0], "cpu",
                       "synthetic", dump_dir=str(tmp_path), dump_every=1)
    assert aee > 0
    files = os.listdir(tmp_path)
    assert any(f.endswith("_pred.flo") for f in files)
    assert any(f.endswith("_pred.jpg") for f in files)
    assert any(f.endswith("_warped.jpg") for f in files)


def test_trainer_action_head(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(64, 96), batch_size=2,
        num_workers=0, model="st_single", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="a", action_classes=5,
        log_interval=1,
    ))
    tr = Trainer(cfg)
    from deepof_amd.data import SyntheticActionDataset

    ds = SyntheticActionDataset(4, 64, 96, num_classes=5)
    batch = {k: v.unsqueeze(0) for k, v in ds[0].items()}
    parts = tr.train_step(batch)
    assert "action_ce" in parts and "unsup" in parts


def test_trainer_augment_step(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(64, 96), batch_size=2,
        num_workers=0, model="flownets", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="g", augment=True, log_interval=1,
    ))
    tr = Trainer(cfg)
    from deepof_amd.data import SyntheticFlowDataset

    ds = SyntheticFlowDataset(4, 64, 96)
    batch = {k: v.unsqueeze(0) for k, v in ds[0].items()}
    parts = tr.train_step(batch)
    assert np.isfinite(parts["total"])


def test_trainer_volume_step(tmp_path):
    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(64, 96), batch_size=1,
        num_workers=0, model="inception_v3", precision="fp32",
        device="cpu", log_dir=str(tmp_path), run_name="v", time_step=3,
        log_interval=1,
    ))
    tr = Trainer(cfg)
    vol = torch.rand(1, 9, 64, 96) * 255
    parts = tr.train_step({"volume": vol})
    assert np.isfinite(parts["total"])


def test_perceptual_warp_loss():
    from deepof_amd.losses import PerceptualWarpLoss

    torch.manual_seed(0)
    fn = PerceptualWarpLoss(levels=(3, 4))  # p2, p1 (cheap on CPU)
    img1 = torch.rand(1, 3, 64, 64)
    img2 = torch.rand(1, 3, 64, 64)
    flow = torch.zeros(1, 2, 64, 64, requires_grad=True)
    loss = fn(flow, img1, img2)
    assert torch.isfinite(loss)
    loss.backward()
    assert torch.isfinite(flow.grad).all()
    # identical images + zero flow -> near-minimal loss vs random flow
    loss0 = fn(torch.zeros(1, 2, 64, 64), img1, img1)
    lossr = fn(torch.randn(1, 2, 64, 64) * 8, img1, img1)
    assert float(loss0) < float(lossr)


def test_trainer_profiler_smoke(tmp_path):
    import os

    from deepof_amd.config import Config
    from deepof_amd.engine import Trainer

    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(32, 48), batch_size=2,
        num_workers=0, model="flownets", precision="fp32", device="cpu",
        log_dir=str(tmp_path), run_name="p", profile_steps=2,
        log_interval=100,
    ))
    Trainer(cfg).fit(max_steps=6)
    files = os.listdir(os.path.join(str(tmp_path), "p"))
    assert any("trace" in f or f.endswith(".json") for f in files), files


def test_edge_aware_smoothness():
    from deepof_amd.ops.reference import (image_gradient_masks,
                                          unsup_loss_scale_edge_aware)

    torch.manual_seed(0)
    img1 = torch.rand(1, 3, 24, 32)
    img2 = torch.rand(1, 3, 24, 32)
    gm = image_gradient_masks(img1)
    assert gm.shape == (1, 2, 24, 32)
    assert (gm >= -1e-5).all() and (gm <= 1.0 + 1e-5).all()
    flow = torch.randn(1, 2, 24, 32, requires_grad=True)
    res = unsup_loss_scale_edge_aware(flow, img1, img2, 1.0)
    res["total"].backward()
    assert torch.isfinite(flow.grad).all()
    # edge-aware smoothness never exceeds the unweighted one
    from deepof_amd.ops import reference as ref

    base = ref.unsup_loss_scale(flow.detach(), img1, img2, 1.0)
    assert float(res["u_loss"]) <= float(base["u_loss"]) + 1e-6


def test_evaluate_accuracy():
    from deepof_amd.data import SyntheticActionDataset, build_dataloader
    from deepof_amd.engine.evaluator import evaluate_accuracy
    from deepof_amd.models import STSingle

    ds = SyntheticActionDataset(4, 64, 96, num_classes=5)
    dl = build_dataloader(ds, 2, shuffle=False, num_workers=0,
                          drop_last=False)
    m = STSingle(input_hw=(64, 96), num_classes=5)
    acc = evaluate_accuracy(m, dl, ds.mean_bgr, "cpu")
    assert 0.0 <= acc <= 1.0


def test_predict_flow_joint_model():
    from deepof_amd.engine.evaluator import predict_flow
    from deepof_amd.models import STSingle

    m = STSingle(input_hw=(64, 96), num_classes=5)
    img1 = torch.rand(1, 3, 64, 96) * 255
    img2 = torch.rand(1, 3, 64, 96) * 255
    pred = predict_flow(m, img1, img2, (104.0, 117.0, 123.0), 10.0,
                        "ucf101", gt_size=(64, 96))
    assert pred.shape == (1, 2, 64, 96)


def test_evaluate_aee_volume():
    from deepof_amd.data import build_dataloader
    from deepof_amd.engine.evaluator import evaluate_aee
    from deepof_amd.models import build_model
    from torch.utils.data import Dataset

    class VolDs(Dataset):
        def __len__(self):
            return 2

        def __getitem__(self, i):
            torch.manual_seed(i)
            return {"volume": torch.rand(9, 64, 96) * 255,
                    "flow": torch.randn(4, 64, 96)}

    model, scales, _ = build_model("inception_v3", time_step=3)
    dl = build_dataloader(VolDs(), 1, shuffle=False, num_workers=0,
                          drop_last=False)
    aee = evaluate_aee(model, dl, (70.0, 83.0, 92.0), scales[0], "cpu",
                       "sintel")
    assert aee > 0


def test_eval_postproc_overrides():
    """Per-config eval overrides (the VGG chairs variant
    clips to [-204.479, 201.3478]; version1 sintel uses a different
    amplifier) flow through predict_flow."""
    import torch

    from deepof_amd.engine.evaluator import _postproc, predict_flow
    from deepof_amd.models import build_model

    assert _postproc("flying_chairs") == (2.0, -300.0, 250.0)
    assert _postproc("flying_chairs", clip=(-204.479, 201.3478)) == \
        (2.0, -204.479, 201.3478)
    assert _postproc("sintel", mult=100.0) == (100.0, -420.621, 426.311)

    torch.manual_seed(0)
    model, scales, _ = build_model("flownets")
    img = torch.rand(1, 3, 64, 96) * 255
    p_default = predict_flow(model, img, img, (127.5,) * 3, scales[0],
                             "flying_chairs")
    p_clip = predict_flow(model, img, img, (127.5,) * 3, scales[0],
                          "flying_chairs", clip=(-0.01, 0.01))
    assert p_clip.abs().max() <= 0.01
    assert p_default.shape == p_clip.shape
