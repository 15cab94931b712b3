import torch

from deepof_amd.losses import (MultiScaleGuidedLoss, MultiScaleUnsupLoss,
                               preprocess_images)
from deepof_amd.models import FlowNetS
from deepof_amd.models.flownet import FLOW_SCALES

W = [16.0, 8.0, 4.0, 2.0, 1.0, 1.0]


def _fake_batch(b=1, h=64, w=96):
    torch.manual_seed(0)
    img1 = torch.rand(b, 3, h, w) * 255
    img2 = torch.rand(b, 3, h, w) * 255
    return img1, img2


def test_unsup_loss_end_to_end():
    img1, img2 = _fake_batch()
    model = FlowNetS()
    loss_fn = MultiScaleUnsupLoss(FLOW_SCALES, W)
    x1 = preprocess_images(img1, loss_fn.mean_bgr)
    x2 = preprocess_images(img2, loss_fn.mean_bgr)
    flows = model(torch.cat([x1, x2], dim=1))
    res = loss_fn(flows, img1, img2, want_recon=True)
    assert torch.isfinite(res["total"])
    assert len(res["scales"]) == 6
    assert res["recon"].shape == (1, 3, 32, 48)
    assert len(res["flows_all"]) == 6
    res["total"].backward()
    gsum = sum(p.grad.abs().sum() for p in model.parameters()
               if p.grad is not None)
    assert torch.isfinite(gsum) and gsum > 0


def test_guided_loss():
    torch.manual_seed(0)
    flows = [torch.randn(1, 2, 32 // (1 << k), 48 // (1 << k),
                         requires_grad=True) for k in range(3)]
    gt = torch.randn(1, 2, 64, 96) * 4
    fn = MultiScaleGuidedLoss([10.0, 5.0, 2.5], [16.0, 8.0, 4.0])
    res = fn(flows, gt)
    assert torch.isfinite(res["total"])
    res["total"].backward()
    assert all(torch.isfinite(f.grad).all() for f in flows)


def test_guided_loss_zero_at_perfect_prediction():
    gt = torch.ones(1, 2, 16, 16) * 8.0
    # prediction at half res with flow_scale 10: raw = gt_scaled / 10
    from deepof_amd.losses.guided import downscale_flow

    gt_half = downscale_flow(gt, 8, 8)
    pred = (gt_half / 10.0).requires_grad_(True)
    fn = MultiScaleGuidedLoss([10.0], [1.0], epsilon=1e-3, alpha=0.4)
    res = fn([pred], gt)
    # loss at perfect prediction ~= (eps^2)^alpha
    assert float(res["total"]) < (1e-3**2) ** 0.4 + 1e-6
