"""UCF101 joint flow + action models.

Parity targets (/root/reference/ucf101wrapFlow.py):
- STSingle (:62-194): one shared VGG16 encoder over the concatenated
  frame pair; a spatial branch (pool5 flatten -> fc6 -> fc7 -> fc8(101))
  for action classification and a temporal decoder (pr5..pr1, 5 scales)
  for unsupervised flow; joint loss = sum(w_i * flow_i) + w0 * CE.
- STBaseline (:197-363): separate FlowNetS temporal stream and VGG16
  spatial stream fused at pool5 via 1x1 conv.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .common import FlowDecoder, conv, init_flow_module, make_act
from .flownet import FlowNetS
from .vgg16 import VGG16Encoder, VGG_FLOW_SCALES


class STSingle(nn.Module):
    """Shared VGG16 encoder -> (flow pyramid, action logits)."""

    num_scales = 5
    flow_scales = VGG_FLOW_SCALES

    def __init__(self, input_hw: tuple[int, int] = (256, 320),
                 num_classes: int = 101, act: str = "elu",
                 fc_dim: int = 4096, dropout: float = 0.5):
        super().__init__()
        self.encoder = VGG16Encoder(6, act="relu")
        self.decoder = FlowDecoder([512, 512, 256, 128, 64],
                                   [256, 128, 64, 32], act=act)
        h, w = input_hw
        flat_dim = (h // 32) * (w // 32) * 512
        self.classifier = nn.Sequential(
            nn.Linear(flat_dim, fc_dim), make_act("relu"), nn.Dropout(dropout),
            nn.Linear(fc_dim, fc_dim), make_act("relu"), nn.Dropout(dropout),
            nn.Linear(fc_dim, num_classes),
        )
        init_flow_module(self)

    def forward(self, x: torch.Tensor):
        """x: [B, 6, H, W] pair. Returns (flows finest-first, logits)."""
        feats = self.encoder(x)
        flows = self.decoder(feats)[::-1]
        logits = self.classifier(feats[0].float().flatten(1))
        return flows, logits


class STBaseline(nn.Module):
    """Separate FlowNetS temporal + VGG16 spatial streams, fused at
    pool5 by a 1x1 conv (ucf101wrapFlow.py:332-337)."""

    num_scales = 6

    def __init__(self, input_hw: tuple[int, int] = (256, 320),
                 num_classes: int = 101, act: str = "elu",
                 fc_dim: int = 4096, dropout: float = 0.5):
        super().__init__()
        self.temporal = FlowNetS(act=act)
        self.spatial = VGG16Encoder(3, act="relu")
        # fuse pool5 (512) with conv5_2 features (512) -> 512
        self.fuse = conv(1024, 512, 1, 1, "relu")
        h, w = input_hw
        flat_dim = (h // 32) * (w // 32) * 512
        self.classifier = nn.Sequential(
            nn.Linear(flat_dim, fc_dim), make_act("relu"), nn.Dropout(dropout),
            nn.Linear(fc_dim, fc_dim), make_act("relu"), nn.Dropout(dropout),
            nn.Linear(fc_dim, num_classes),
        )
        init_flow_module(self.fuse)

    def forward(self, x: torch.Tensor):
        im1 = x[:, :3]
        feats = self.temporal.encode(x)       # [c6..c1]
        flows = self.temporal.decoder(feats)[::-1]
        p5 = self.spatial(im1)[0]             # 1/32, 512
        t5 = feats[1]                         # temporal conv5_2, 1/32, 512
        if t5.shape[-2:] != p5.shape[-2:]:
            t5 = torch.nn.functional.adaptive_avg_pool2d(t5, p5.shape[-2:])
        fused = self.fuse(torch.cat([p5, t5], dim=1))
        logits = self.classifier(fused.float().flatten(1))
        return flows, logits
