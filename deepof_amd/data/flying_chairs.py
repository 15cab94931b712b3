"""FlyingChairs dataset.

Split protocol parity (/root/reference/flyingChairsLoader.py:30-62):
a split file with one label per sample line — 1 = train (22,232),
2 = val (640) — over samples named NNNNN_img1.ppm / NNNNN_img2.ppm /
NNNNN_flow.flo (NNNNN is 1-based, %05d).  Images are resized to
image_size; ground-truth flow is loaded at native resolution and used
for eval AEE only (training is unsupervised; flyingChairsTrain.py:173).

BGR mean: (97.533, 99.238, 97.056) (flyingChairsWrapFlow.py:16).
"""

from __future__ import annotations

import os

import numpy as np
import torch
from torch.utils.data import Dataset

from ..utils import read_flo
from .image import load_image, to_chw

FLYING_CHAIRS_MEAN_BGR = (97.533268, 99.238236, 97.055973)


class FlyingChairsDataset(Dataset):
    def __init__(self, data_dir: str, split: str = "train",
                 split_file: str | None = None,
                 image_size: tuple[int, int] | None = (384, 512),
                 load_flow: bool = True):
        assert split in ("train", "val")
        self.data_dir = data_dir
        self.image_size = image_size
        self.load_flow = load_flow
        self.mean_bgr = FLYING_CHAIRS_MEAN_BGR

        if split_file is None:
            split_file = os.path.join(data_dir, "FlyingChairs_train_val.txt")
        with open(split_file) as f:
            labels = [int(line.strip()) for line in f if line.strip()]
        want = 1 if split == "train" else 2
        self.ids = [i + 1 for i, lab in enumerate(labels) if lab == want]
        if not self.ids:
            raise RuntimeError(f"empty {split} split from {split_file}")

    def __len__(self):
        return len(self.ids)

    def _paths(self, sample_id: int):
        p = os.path.join(self.data_dir, f"{sample_id:05d}")
        return f"{p}_img1.ppm", f"{p}_img2.ppm", f"{p}_flow.flo"

    def __getitem__(self, idx):
        sid = self.ids[idx]
        p1, p2, pf = self._paths(sid)
        img1 = to_chw(load_image(p1, self.image_size))
        img2 = to_chw(load_image(p2, self.image_size))
        item = {
            "img1": torch.from_numpy(img1),
            "img2": torch.from_numpy(img2),
        }
        if self.load_flow:
            flow = read_flo(pf)  # [H, W, 2] native resolution (for AEE)
            item["flow"] = torch.from_numpy(
                np.ascontiguousarray(flow.transpose(2, 0, 1))
            )
        return item
