"""VGG feature-space (perceptual) warp loss.

Capability target: the "VGG perceptual loss" configuration of
BASELINE.json configs[4].  Photometric consistency is measured in VGG16
feature space instead of (or on top of) pixel space: the second frame's
features are backward-warped by the (downscaled) predicted flow and
penalized against the first frame's features with a Charbonnier.
Unlike the pixel loss, gradients here flow to BOTH the flow and the
feature extractor (the standalone HIP warp kernel provides d(features)
via atomic scatter and d(flow) analytically).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..models.vgg16 import VGG16Encoder
from .guided import downscale_flow


class PerceptualWarpLoss(nn.Module):
    def __init__(self, levels=(0, 1, 2), epsilon: float = 1e-3,
                 alpha: float = 0.4, train_encoder: bool = False,
                 in_channels: int = 3):
        super().__init__()
        self.encoder = VGG16Encoder(in_channels, act="relu")
        self.levels = levels  # indices into [p5, p4, p3, p2, p1]
        self.epsilon = epsilon
        self.alpha = alpha
        if not train_encoder:
            for p in self.encoder.parameters():
                p.requires_grad_(False)

    def forward(self, flow_finest: torch.Tensor, img1_norm: torch.Tensor,
                img2_norm: torch.Tensor):
        """flow_finest: [B,2,h,w] in PIXELS at full image resolution
        scale; img*_norm: normalized [B,3,H,W] images."""
        f1 = self.encoder(img1_norm.float())[::-1]  # finest first: p1..p5
        f2 = self.encoder(img2_norm.float())[::-1]
        total = None
        for li in self.levels:
            a, b = f1[li], f2[li]
            h, w = a.shape[-2:]
            fl = downscale_flow(flow_finest.float(), h, w)
            recon = ops.warp_bilinear(b, fl)
            d2 = ((recon - a) ** 2).sum(dim=1)
            ew = torch.pow(d2 + self.epsilon**2, self.alpha)
            term = ew.mean()
            total = term if total is None else total + term
        return total / len(self.levels)
