"""Inception-v3 encoder + 6-scale flow decoder (multi-frame capable).

Parity target: /root/reference/sintelWrapFlow.py:5-340 (inception_v3_base
Conv2d_1a..Mixed_7c) and its decoder :386-443 — skips from Mixed_7c
(1/32), Mixed_6e (1/16), Mixed_5d (1/8), MaxPool_5a (1/8), MaxPool_3a
(1/4), Conv2d_1a (1/2), with a stride-1 refinement between the two
1/8-resolution skips and flow scales [10, 5, 2.5, 2.5, 1.25, 0.625]
finest-first.  The v0 chairs variant is
/root/reference/flyingChairsWrapFlow.py:131-633.

Multi-frame Sintel volumes: in_channels = 3*T, flow_channels = 2*(T-1)
(sintelWrapFlow.py:351-352).  Unlike the reference (whose up_pr deconvs
emit 2 channels even when T > 2 — a bug), the flow-upsample path carries
all 2*(T-1) channels.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .common import FlowDecoder, init_flow_module

# finest (pr1 at 1/2) -> coarsest (pr6 at 1/32); two 1/8 scales
INCEPTION_FLOW_SCALES = [10.0, 5.0, 2.5, 2.5, 1.25, 0.625]


class BasicConv2d(nn.Module):
    def __init__(self, cin, cout, k, stride=1, padding=None):
        super().__init__()
        if padding is None:
            padding = tuple(x // 2 for x in k) if isinstance(k, tuple) else k // 2
        self.conv = nn.Conv2d(cin, cout, k, stride=stride, padding=padding,
                              bias=False)
        self.bn = nn.BatchNorm2d(cout, eps=0.001)
        self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.act(self.bn(self.conv(x)))


class InceptionA(nn.Module):
    def __init__(self, cin, pool_features):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 64, 1)
        self.b5x5 = nn.Sequential(BasicConv2d(cin, 48, 1),
                                  BasicConv2d(48, 64, 5))
        self.b3x3dbl = nn.Sequential(BasicConv2d(cin, 64, 1),
                                     BasicConv2d(64, 96, 3),
                                     BasicConv2d(96, 96, 3))
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  BasicConv2d(cin, pool_features, 1))

    def forward(self, x):
        return torch.cat([self.b1x1(x), self.b5x5(x), self.b3x3dbl(x),
                          self.pool(x)], 1)


class InceptionB(nn.Module):  # reduction, stride 2 (Mixed_6a)
    def __init__(self, cin):
        super().__init__()
        self.b3x3 = BasicConv2d(cin, 384, 3, stride=2)
        self.b3x3dbl = nn.Sequential(BasicConv2d(cin, 64, 1),
                                     BasicConv2d(64, 96, 3),
                                     BasicConv2d(96, 96, 3, stride=2))
        self.pool = nn.MaxPool2d(3, 2, 1)

    def forward(self, x):
        return torch.cat([self.b3x3(x), self.b3x3dbl(x), self.pool(x)], 1)


class InceptionC(nn.Module):  # 7x1/1x7 factorized (Mixed_6b..6e)
    def __init__(self, cin, c7):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 192, 1)
        self.b7x7 = nn.Sequential(
            BasicConv2d(cin, c7, 1),
            BasicConv2d(c7, c7, (1, 7)),
            BasicConv2d(c7, 192, (7, 1)),
        )
        self.b7x7dbl = nn.Sequential(
            BasicConv2d(cin, c7, 1),
            BasicConv2d(c7, c7, (7, 1)),
            BasicConv2d(c7, c7, (1, 7)),
            BasicConv2d(c7, c7, (7, 1)),
            BasicConv2d(c7, 192, (1, 7)),
        )
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  BasicConv2d(cin, 192, 1))

    def forward(self, x):
        return torch.cat([self.b1x1(x), self.b7x7(x), self.b7x7dbl(x),
                          self.pool(x)], 1)


class InceptionD(nn.Module):  # reduction, stride 2 (Mixed_7a)
    def __init__(self, cin):
        super().__init__()
        self.b3x3 = nn.Sequential(BasicConv2d(cin, 192, 1),
                                  BasicConv2d(192, 320, 3, stride=2))
        self.b7x7x3 = nn.Sequential(
            BasicConv2d(cin, 192, 1),
            BasicConv2d(192, 192, (1, 7)),
            BasicConv2d(192, 192, (7, 1)),
            BasicConv2d(192, 192, 3, stride=2),
        )
        self.pool = nn.MaxPool2d(3, 2, 1)

    def forward(self, x):
        return torch.cat([self.b3x3(x), self.b7x7x3(x), self.pool(x)], 1)


class InceptionE(nn.Module):  # Mixed_7b / 7c
    def __init__(self, cin):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 320, 1)
        self.b3x3_1 = BasicConv2d(cin, 384, 1)
        self.b3x3_2a = BasicConv2d(384, 384, (1, 3))
        self.b3x3_2b = BasicConv2d(384, 384, (3, 1))
        self.b3x3dbl_1 = nn.Sequential(BasicConv2d(cin, 448, 1),
                                       BasicConv2d(448, 384, 3))
        self.b3x3dbl_2a = BasicConv2d(384, 384, (1, 3))
        self.b3x3dbl_2b = BasicConv2d(384, 384, (3, 1))
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  BasicConv2d(cin, 192, 1))

    def forward(self, x):
        a = self.b3x3_1(x)
        b = self.b3x3dbl_1(x)
        return torch.cat([
            self.b1x1(x),
            torch.cat([self.b3x3_2a(a), self.b3x3_2b(a)], 1),
            torch.cat([self.b3x3dbl_2a(b), self.b3x3dbl_2b(b)], 1),
            self.pool(x),
        ], 1)


class InceptionV3Base(nn.Module):
    """Conv2d_1a .. Mixed_7c with SAME-style padding; returns the six
    skip endpoints, coarsest first."""

    def __init__(self, in_channels: int = 6):
        super().__init__()
        self.conv1a = BasicConv2d(in_channels, 32, 3, stride=2)  # 1/2
        self.conv2a = BasicConv2d(32, 32, 3)
        self.conv2b = BasicConv2d(32, 64, 3)
        self.pool3a = nn.MaxPool2d(3, 2, 1)                      # 1/4
        self.conv3b = BasicConv2d(64, 80, 1)
        self.conv4a = BasicConv2d(80, 192, 3)
        self.pool5a = nn.MaxPool2d(3, 2, 1)                      # 1/8
        self.mixed5b = InceptionA(192, 32)   # 256
        self.mixed5c = InceptionA(256, 64)   # 288
        self.mixed5d = InceptionA(288, 64)   # 288
        self.mixed6a = InceptionB(288)       # 768, 1/16
        self.mixed6b = InceptionC(768, 128)
        self.mixed6c = InceptionC(768, 160)
        self.mixed6d = InceptionC(768, 160)
        self.mixed6e = InceptionC(768, 192)
        self.mixed7a = InceptionD(768)       # 1280, 1/32
        self.mixed7b = InceptionE(1280)      # 2048
        self.mixed7c = InceptionE(2048)      # 2048

    def forward(self, x):
        c1a = self.conv1a(x)
        p3a = self.pool3a(self.conv2b(self.conv2a(c1a)))
        p5a = self.pool5a(self.conv4a(self.conv3b(p3a)))
        m5d = self.mixed5d(self.mixed5c(self.mixed5b(p5a)))
        m6e = self.mixed6e(self.mixed6d(self.mixed6c(self.mixed6b(
            self.mixed6a(m5d)))))
        m7c = self.mixed7c(self.mixed7b(self.mixed7a(m6e)))
        return [m7c, m6e, m5d, p5a, p3a, c1a]


class InceptionFlow(nn.Module):
    """Inception-v3 flow model; supports multi-frame channel-stacked
    volumes (time_step > 2).  Finest scale first in the output."""

    num_scales = 6
    flow_scales = INCEPTION_FLOW_SCALES

    def __init__(self, act: str = "elu", time_step: int = 2):
        super().__init__()
        self.time_step = time_step
        flow_channels = 2 * (time_step - 1)
        self.encoder = InceptionV3Base(in_channels=3 * time_step)
        self.decoder = FlowDecoder(
            [2048, 768, 288, 192, 64, 32],
            [512, 256, 128, 64, 32],
            act=act,
            flow_channels=flow_channels,
            up_factors=[2, 2, 1, 2, 2],
        )
        init_flow_module(self)

    def forward(self, x: torch.Tensor) -> list[torch.Tensor]:
        feats = self.encoder(x)
        return self.decoder(feats)[::-1]
