"""Synthetic datasets for benchmarks and tests (no network, no real data).

SyntheticFlowDataset generates a smooth random flow field, renders img1
as band-limited random texture and img2 as img1 forward-displaced by the
flow (nearest splat with fill), so the unsupervised photometric loss has
real signal and AEE against the known flow is meaningful.
"""

from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import Dataset


def _smooth_noise(rng, h, w, cells=8, channels=1):
    """Band-limited noise: bilinear-upsampled coarse random grid."""
    coarse = rng.standard_normal((channels, cells + 1, cells + 1)).astype(np.float32)
    t = torch.from_numpy(coarse).unsqueeze(0)
    up = torch.nn.functional.interpolate(
        t, size=(h, w), mode="bilinear", align_corners=True
    )
    return up[0].numpy()


def make_synthetic_pair(rng, h, w, max_flow=8.0):
    """Returns (img1, img2, flow): img* [3,H,W] 0-255 BGR, flow [2,H,W].

    img1 is DEFINED as the bilinear backward warp of img2 by the flow —
    exactly the model the unsupervised loss assumes — so the ground
    truth flow is a true minimum of the photometric term.
    """
    from ..ops.reference import warp_bilinear

    tex = _smooth_noise(rng, h, w, cells=24, channels=3)
    img2 = (tex - tex.min()) / (np.ptp(tex) + 1e-6) * 255.0

    flow = _smooth_noise(rng, h, w, cells=6, channels=2) * (max_flow / 2.0)
    flow = np.clip(flow, -max_flow, max_flow).astype(np.float32)

    with torch.no_grad():
        img1 = warp_bilinear(
            torch.from_numpy(img2.astype(np.float32)).unsqueeze(0),
            torch.from_numpy(flow).unsqueeze(0),
        )[0].numpy()
    return img1.astype(np.float32), img2.astype(np.float32), flow


class SyntheticFlowDataset(Dataset):
    def __init__(self, num_samples: int, height: int, width: int,
                 max_flow: float = 8.0, seed: int = 0):
        self.num_samples = num_samples
        self.h, self.w = height, width
        self.max_flow = max_flow
        self.seed = seed
        self.mean_bgr = (127.5, 127.5, 127.5)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.default_rng(self.seed * 1000003 + idx)
        img1, img2, flow = make_synthetic_pair(rng, self.h, self.w,
                                               self.max_flow)
        return {
            "img1": torch.from_numpy(img1),
            "img2": torch.from_numpy(img2),
            "flow": torch.from_numpy(flow),
        }


class SyntheticActionDataset(SyntheticFlowDataset):
    """Adds a class label (for the UCF101 joint flow+action config)."""

    def __init__(self, num_samples, height, width, num_classes=101, **kw):
        super().__init__(num_samples, height, width, **kw)
        self.num_classes = num_classes

    def __getitem__(self, idx):
        item = super().__getitem__(idx)
        item["label"] = torch.tensor(idx % self.num_classes, dtype=torch.long)
        return item
