"""VGG16 warm start from the classic vgg16_weights.npz.

Parity: /root/reference/flyingChairsTrain.py:60-76 and
ucf101train.py:68-88 — assign conv1_1..conv5_3 weights from the npz;
for 6-channel inputs the conv1_1 kernel is repeated across the two
stacked frames (and halved so the response scale is preserved).
"""

from __future__ import annotations

import numpy as np
import torch

# npz keys are conv1_1_W, conv1_1_b, ... in [R, S, C, K] (TF) order
_VGG_LAYERS = [
    ("block1", ["conv1_1", "conv1_2"]),
    ("block2", ["conv2_1", "conv2_2"]),
    ("block3", ["conv3_1", "conv3_2", "conv3_3"]),
    ("block4", ["conv4_1", "conv4_2", "conv4_3"]),
    ("block5", ["conv5_1", "conv5_2", "conv5_3"]),
]


def load_vgg16_npz(encoder, npz_path: str) -> int:
    """Copy npz weights into a VGG16Encoder; returns #tensors loaded."""
    data = np.load(npz_path)
    loaded = 0
    with torch.no_grad():
        for block_name, layer_names in _VGG_LAYERS:
            block = getattr(encoder, block_name)
            convs = [m for m in block.modules()
                     if isinstance(m, torch.nn.Conv2d)]
            assert len(convs) == len(layer_names), (block_name, len(convs))
            for conv, lname in zip(convs, layer_names):
                wk, bk = f"{lname}_W", f"{lname}_b"
                if wk not in data:
                    continue
                w = torch.from_numpy(data[wk]).permute(3, 2, 0, 1)  # KCRS
                if conv.in_channels == 2 * w.shape[1]:
                    # 6-channel input: tile across both frames, halve
                    w = torch.cat([w, w], dim=1) * 0.5
                if tuple(w.shape) != tuple(conv.weight.shape):
                    raise ValueError(
                        f"{lname}: npz {tuple(w.shape)} vs model "
                        f"{tuple(conv.weight.shape)}")
                conv.weight.copy_(w.to(conv.weight.dtype))
                conv.bias.copy_(torch.from_numpy(data[bk]).to(conv.bias.dtype))
                loaded += 2
    return loaded
