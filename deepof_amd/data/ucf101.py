"""UCF101 frame-pair dataset (joint flow + action training).

Parity (/root/reference/ucf101Loader.py:27-87): walks
frames/<class>/<clip>/ directories; the clip's group number g (from the
vXX_gYY_cZZ naming) splits train (g > 7) vs test; training samples are
class-balanced random (frame, frame+1) pairs with the class label.
"""

from __future__ import annotations

import os
import re

import numpy as np
import torch
from torch.utils.data import Dataset

from .image import load_image, to_chw

UCF101_MEAN_BGR = (104.0, 117.0, 123.0)
_GROUP_RE = re.compile(r"_g(\d+)_")


class UCF101Dataset(Dataset):
    def __init__(self, data_dir: str, split: str = "train",
                 image_size: tuple[int, int] | None = (256, 320),
                 group_threshold: int = 7, seed: int = 0):
        assert split in ("train", "test")
        self.image_size = image_size
        self.mean_bgr = UCF101_MEAN_BGR
        frames_root = os.path.join(data_dir, "frames")
        self.classes = sorted(
            d for d in os.listdir(frames_root)
            if os.path.isdir(os.path.join(frames_root, d))
        )
        self.class_to_idx = {c: i for i, c in enumerate(self.classes)}

        self.clips: list[tuple[str, int]] = []  # (clip_dir, label)
        for c in self.classes:
            cdir = os.path.join(frames_root, c)
            for clip in sorted(os.listdir(cdir)):
                clip_dir = os.path.join(cdir, clip)
                if not os.path.isdir(clip_dir):
                    continue
                m = _GROUP_RE.search(clip)
                group = int(m.group(1)) if m else 0
                is_train = group > group_threshold
                if (split == "train") == is_train:
                    self.clips.append((clip_dir, self.class_to_idx[c]))
        if not self.clips:
            raise RuntimeError(f"no {split} clips under {frames_root}")
        self.seed = seed
        # per-clip draw counters: the frame pick is derived from
        # (seed, idx, visit-count) so repeated visits to one clip
        # (across epochs) decorrelate, and DataLoader workers — which
        # each hold a copy of this dataset but serve disjoint indices —
        # never replay each other's sequences (one shared default_rng
        # copied into every persistent worker did exactly that).
        self._visits = np.zeros(len(self.clips), dtype=np.int64)

    def __len__(self):
        return len(self.clips)

    def __getitem__(self, idx):
        clip_dir, label = self.clips[idx]
        frames = sorted(
            f for f in os.listdir(clip_dir)
            if f.lower().endswith((".jpg", ".jpeg", ".png"))
        )
        if len(frames) < 2:
            raise RuntimeError(f"clip {clip_dir} has <2 frames")
        rng = np.random.default_rng((self.seed, idx, int(self._visits[idx])))
        self._visits[idx] += 1
        i = int(rng.integers(0, len(frames) - 1))
        img1 = to_chw(load_image(os.path.join(clip_dir, frames[i]),
                                 self.image_size))
        img2 = to_chw(load_image(os.path.join(clip_dir, frames[i + 1]),
                                 self.image_size))
        return {
            "img1": torch.from_numpy(img1),
            "img2": torch.from_numpy(img2),
            "label": torch.tensor(label, dtype=torch.long),
        }
