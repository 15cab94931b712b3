"""FlowNetS and FlowNetC encoders with the shared 6-scale flow decoder.

FlowNetS parity: the contracting stack of
/root/reference/flyingChairsWrapFlow.py:31-40 (conv1 7x7/2 6->64 ...
conv6_2 3x3 1024, ELU) and its expanding part (:58-119).  Raw flow
predictions pr_k carry the per-scale flow_scale 20/2^k applied by the
loss/eval layers (flow scales at :63-118).

FlowNetC: not in the reference TF repo — a BASELINE.json configs[2]
requirement.  Siamese conv1..conv3 towers, correlation cost volume
(max displacement 10 -> 441 channels) via the hand-written HIP kernel,
conv_redir 1x1, then the FlowNetS tail.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import correlation
from .common import FlowDecoder, conv, init_flow_module

# flow_scale at pyramid level k (pr1 finest at stride 2): 20 / 2^k
FLOW_SCALES = [20.0 / (1 << k) for k in range(1, 7)]  # [10, 5, 2.5, 1.25, .625, .3125]


class FlowNetS(nn.Module):
    num_scales = 6
    # finest -> coarsest, matching FLOW_SCALES order after reversal
    def __init__(self, act: str = "elu", in_channels: int = 6,
                 flow_channels: int = 2):
        super().__init__()
        self.conv1 = conv(in_channels, 64, 7, 2, act)
        self.conv2 = conv(64, 128, 5, 2, act)
        self.conv3_1 = conv(128, 256, 5, 2, act)
        self.conv3_2 = conv(256, 256, 3, 1, act)
        self.conv4_1 = conv(256, 512, 3, 2, act)
        self.conv4_2 = conv(512, 512, 3, 1, act)
        self.conv5_1 = conv(512, 512, 3, 2, act)
        self.conv5_2 = conv(512, 512, 3, 1, act)
        self.conv6_1 = conv(512, 1024, 3, 2, act)
        self.conv6_2 = conv(1024, 1024, 3, 1, act)
        self.decoder = FlowDecoder(
            [1024, 512, 512, 256, 128, 64], [512, 256, 128, 64, 32],
            act=act, flow_channels=flow_channels,
        )
        init_flow_module(self)

    def encode(self, x: torch.Tensor) -> list[torch.Tensor]:
        """Contracting stack; returns [c6, c5, c4, c3, c2, c1]."""
        c1 = self.conv1(x)
        c2 = self.conv2(c1)
        c3 = self.conv3_2(self.conv3_1(c2))
        c4 = self.conv4_2(self.conv4_1(c3))
        c5 = self.conv5_2(self.conv5_1(c4))
        c6 = self.conv6_2(self.conv6_1(c5))
        return [c6, c5, c4, c3, c2, c1]

    def forward(self, x: torch.Tensor) -> list[torch.Tensor]:
        """x: [B, 6, H, W] (concat of the normalized image pair).

        Returns raw flow predictions FINEST FIRST: [pr1 ... pr6]
        (pr1 at H/2 x W/2), to be scaled by FLOW_SCALES[k].
        """
        return self.decoder(self.encode(x))[::-1]


class FlowNetC(nn.Module):
    num_scales = 6

    def __init__(self, act: str = "elu", max_displacement: int = 10):
        super().__init__()
        self.md = max_displacement
        self.conv1 = conv(3, 64, 7, 2, act)
        self.conv2 = conv(64, 128, 5, 2, act)
        self.conv3 = conv(128, 256, 5, 2, act)
        self.conv_redir = conv(256, 32, 1, 1, act)
        corr_ch = (2 * max_displacement + 1) ** 2
        self.conv3_1 = conv(corr_ch + 32, 256, 3, 1, act)
        self.conv4_1 = conv(256, 512, 3, 2, act)
        self.conv4_2 = conv(512, 512, 3, 1, act)
        self.conv5_1 = conv(512, 512, 3, 2, act)
        self.conv5_2 = conv(512, 512, 3, 1, act)
        self.conv6_1 = conv(512, 1024, 3, 2, act)
        self.conv6_2 = conv(1024, 1024, 3, 1, act)
        self.decoder = FlowDecoder(
            [1024, 512, 512, 256, 128, 64], [512, 256, 128, 64, 32], act=act
        )
        init_flow_module(self)

    def forward(self, x: torch.Tensor) -> list[torch.Tensor]:
        """x: [B, 6, H, W]; the two images run the siamese tower."""
        im1, im2 = x[:, :3], x[:, 3:]
        c1a = self.conv1(im1)
        c2a = self.conv2(c1a)
        c3a = self.conv3(c2a)
        c1b = self.conv1(im2)
        c2b = self.conv2(c1b)
        c3b = self.conv3(c2b)
        # bf16 inputs, fp32 accumulation inside the HIP kernel
        corr = correlation(c3a, c3b, self.md).to(c3a.dtype)
        redir = self.conv_redir(c3a)
        c3 = self.conv3_1(torch.cat([corr, redir], dim=1))
        c4 = self.conv4_2(self.conv4_1(c3))
        c5 = self.conv5_2(self.conv5_1(c4))
        c6 = self.conv6_2(self.conv6_1(c5))
        flows_coarse_first = self.decoder([c6, c5, c4, c3, c2a, c1a])
        return flows_coarse_first[::-1]
