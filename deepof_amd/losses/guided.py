"""Guided (proxy-label) supervision.

The reference TF repo trains only the unsupervised photometric baseline
(ground-truth .flo is loaded solely for eval AEE —
/root/reference/flyingChairsTrain.py:173); the paper's guided term
(supervise against proxy flow labels, then photometric fine-tune) lives
in its Caffe/PyTorch siblings and is required capability here
(BASELINE.json north_star).

Per scale k: downsample the (proxy) label to the prediction's
resolution, rescale magnitudes by the resolution ratio, and penalize the
endpoint difference with a Charbonnier: ((du^2 + dv^2) + eps^2)^alpha,
averaged over pixels, weighted by the same per-scale weights.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


def downscale_flow(flow_gt: torch.Tensor, h: int, w: int) -> torch.Tensor:
    """Bilinear-resize a [B,2,H,W] flow field and rescale magnitudes."""
    b, _, H, W = flow_gt.shape
    f = ops.resize_bilinear(flow_gt, h, w)
    scale = torch.tensor([w / W, h / H], dtype=f.dtype, device=f.device)
    return f * scale.view(1, 2, 1, 1)


class MultiScaleGuidedLoss(nn.Module):
    def __init__(self, flow_scales: list[float], loss_weights: list[float],
                 epsilon: float = 1e-3, alpha: float = 0.4):
        super().__init__()
        self.flow_scales = flow_scales
        self.loss_weights = loss_weights
        self.epsilon = epsilon
        self.alpha = alpha

    def forward(self, flows: list[torch.Tensor], flow_gt: torch.Tensor):
        """flows: raw predictions finest first; flow_gt: [B,2,H,W] pixels
        at full resolution (proxy or true labels)."""
        total = None
        scale_losses = []
        gt = flow_gt.float().detach()
        for k, flow in enumerate(flows):
            h, w = flow.shape[-2:]
            gt_k = downscale_flow(gt, h, w)
            pred = flow.float() * self.flow_scales[k]
            d = pred - gt_k
            ew = torch.pow(d[:, 0] ** 2 + d[:, 1] ** 2 + self.epsilon**2,
                           self.alpha)
            loss_k = ew.mean()
            scale_losses.append(loss_k)
            term = self.loss_weights[k] * loss_k
            total = term if total is None else total + term
        return {"total": total, "scales": scale_losses}
