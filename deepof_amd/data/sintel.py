"""MPI-Sintel dataset: pair mode and multi-frame volume mode.

Pair mode parity (/root/reference/version1/loader/sintelLoader.py:18-71):
consecutive frame pairs from training/<clean|final>/<scene>/, flow from
training/flow/<scene>/frame_NNNN.flo, split by Sintel_train_val.txt
(1 = train, 2 = val).

Volume mode parity (/root/reference/sintelLoader.py:31-93): sliding
windows of time_step frames per scene; item = frames [3T, H, W] plus
flows [2(T-1), H, W] (one flow per consecutive pair, eval only).
"""

from __future__ import annotations

import os

import numpy as np
import torch
from torch.utils.data import Dataset

from ..utils import read_flo
from .image import load_image, to_chw

SINTEL_MEAN_BGR = (70.1433, 83.1915, 92.8827)


def _scene_frames(img_root: str) -> dict[str, list[str]]:
    scenes = {}
    for scene in sorted(os.listdir(img_root)):
        d = os.path.join(img_root, scene)
        if os.path.isdir(d):
            scenes[scene] = sorted(
                os.path.join(d, f) for f in os.listdir(d) if f.endswith(".png")
            )
    return scenes


class SintelDataset(Dataset):
    def __init__(self, data_dir: str, split: str = "train",
                 pass_key: str = "clean", time_step: int = 2,
                 image_size: tuple[int, int] | None = (436, 1024),
                 split_file: str | None = None, load_flow: bool = True):
        assert split in ("train", "val")
        assert time_step >= 2
        self.data_dir = data_dir
        self.time_step = time_step
        self.image_size = image_size
        self.load_flow = load_flow
        self.mean_bgr = SINTEL_MEAN_BGR

        img_root = os.path.join(data_dir, "training", pass_key)
        self.flow_root = os.path.join(data_dir, "training", "flow")
        scenes = _scene_frames(img_root)

        # windows of time_step consecutive frames within one scene
        self.windows: list[list[str]] = []
        for scene, frames in scenes.items():
            for i in range(len(frames) - time_step + 1):
                self.windows.append(frames[i : i + time_step])

        if split_file is None:
            cand = os.path.join(data_dir, "Sintel_train_val.txt")
            split_file = cand if os.path.exists(cand) else None
        if split_file is not None:
            with open(split_file) as f:
                labels = [int(x.strip()) for x in f if x.strip()]
            want = 1 if split == "train" else 2
            self.windows = [w for w, lab in zip(self.windows, labels)
                            if lab == want]
        elif split == "val":
            # fallback: first window of each scene (reference's
            # 1-window-per-scene val, sintelLoader.py:47-70)
            seen, val = set(), []
            for w in self.windows:
                scene = os.path.basename(os.path.dirname(w[0]))
                if scene not in seen:
                    seen.add(scene)
                    val.append(w)
            self.windows = val
        if not self.windows:
            raise RuntimeError(f"empty {split} split under {img_root}")

    def __len__(self):
        return len(self.windows)

    def _flow_path(self, frame_path: str) -> str:
        scene = os.path.basename(os.path.dirname(frame_path))
        name = os.path.splitext(os.path.basename(frame_path))[0] + ".flo"
        return os.path.join(self.flow_root, scene, name)

    def __getitem__(self, idx):
        frames = self.windows[idx]
        imgs = [to_chw(load_image(p, self.image_size)) for p in frames]
        item = {}
        if self.time_step == 2:
            item["img1"] = torch.from_numpy(imgs[0])
            item["img2"] = torch.from_numpy(imgs[1])
        else:
            item["volume"] = torch.from_numpy(np.concatenate(imgs, axis=0))
        if self.load_flow:
            flows = []
            for p in frames[:-1]:
                f = read_flo(self._flow_path(p))
                flows.append(np.ascontiguousarray(f.transpose(2, 0, 1)))
            flow = np.concatenate(flows, axis=0)
            item["flow"] = torch.from_numpy(flow)
        return item
