"""VGG16 encoder + 5-scale flow decoder.

Parity target: the VGG16 flow model of
/root/reference/version1/model/VGG16.py:22-159 (VGG16 conv stack on the
6-channel image pair, 5-scale decoder, weight_L [7,5,3,3,1]) and the v0
variant /root/reference/flyingChairsWrapFlow_vgg.py.  Also used as the
spatial-stream classifier for UCF101 (ucf101wrapFlow.py:7-60).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .common import FlowDecoder, conv, init_flow_module, make_act

# flow scale at level k (pr1 at 1/2 resolution): 20 / 2^k
VGG_FLOW_SCALES = [20.0 / (1 << k) for k in range(1, 6)]  # 5 scales


class VGG16Encoder(nn.Module):
    """conv1_1..conv5_3 with 2x2 max pools; returns the 5 pool outputs."""

    def __init__(self, in_channels: int = 6, act: str = "relu"):
        super().__init__()

        def block(cin, cout, n):
            layers = []
            for i in range(n):
                layers.append(conv(cin if i == 0 else cout, cout, 3, 1, act))
            return nn.Sequential(*layers)

        self.block1 = block(in_channels, 64, 2)
        self.block2 = block(64, 128, 2)
        self.block3 = block(128, 256, 3)
        self.block4 = block(256, 512, 3)
        self.block5 = block(512, 512, 3)
        self.pool = nn.MaxPool2d(2, 2)

    def forward(self, x):
        p1 = self.pool(self.block1(x))        # 1/2, 64
        p2 = self.pool(self.block2(p1))       # 1/4, 128
        p3 = self.pool(self.block3(p2))       # 1/8, 256
        p4 = self.pool(self.block4(p3))       # 1/16, 512
        p5 = self.pool(self.block5(p4))       # 1/32, 512
        return [p5, p4, p3, p2, p1]


class VGG16Flow(nn.Module):
    """VGG16 flow model: 5 pyramid scales, finest (pr1, 1/2 res) first."""

    num_scales = 5

    def __init__(self, act: str = "elu", in_channels: int = 6):
        super().__init__()
        # The reference keeps ReLU inside the VGG stack and ELU for the
        # decoder (ucf101wrapFlow.py uses ReLU for the classifier path).
        self.encoder = VGG16Encoder(in_channels, act="relu")
        self.decoder = FlowDecoder([512, 512, 256, 128, 64],
                                   [256, 128, 64, 32], act=act)
        init_flow_module(self)

    def forward(self, x: torch.Tensor) -> list[torch.Tensor]:
        feats = self.encoder(x)
        return self.decoder(feats)[::-1]


class VGG16Classifier(nn.Module):
    """VGG16 action classifier head (fc6/fc7/fc8 + softmax-CE is applied
    by the loss): parity with ucf101wrapFlow.py:109-119."""

    def __init__(self, input_hw: tuple[int, int], num_classes: int = 101,
                 in_channels: int = 3, fc_dim: int = 4096,
                 dropout: float = 0.5):
        super().__init__()
        self.encoder = VGG16Encoder(in_channels, act="relu")
        h, w = input_hw
        flat_dim = (h // 32) * (w // 32) * 512
        self.head = nn.Sequential(
            nn.Linear(flat_dim, fc_dim), make_act("relu"),
            nn.Dropout(dropout),
            nn.Linear(fc_dim, fc_dim), make_act("relu"),
            nn.Dropout(dropout),
            nn.Linear(fc_dim, num_classes),
        )

    def forward(self, x):
        p5 = self.encoder(x)[0]
        return self.head(p5.flatten(1))
