"""Loader protocol tests against synthetic on-disk dataset trees."""

import numpy as np
import pytest
import torch
from PIL import Image

from deepof_amd.utils import write_flo


def _save_img(path, h=32, w=48):
    rng = np.random.default_rng(hash(str(path)) % 2**32)
    arr = rng.integers(0, 255, (h, w, 3), dtype=np.uint8)
    Image.fromarray(arr).save(path)


def test_flying_chairs_loader(tmp_path):
    from deepof_amd.data import FlyingChairsDataset

    labels = [1, 1, 2, 1]  # 3 train, 1 val
    (tmp_path / "FlyingChairs_train_val.txt").write_text(
        "\n".join(map(str, labels)))
    for i in range(1, 5):
        _save_img(tmp_path / f"{i:05d}_img1.ppm")
        _save_img(tmp_path / f"{i:05d}_img2.ppm")
        write_flo(tmp_path / f"{i:05d}_flow.flo",
                  np.random.randn(32, 48, 2).astype(np.float32))

    train = FlyingChairsDataset(str(tmp_path), "train", image_size=(32, 48))
    val = FlyingChairsDataset(str(tmp_path), "val", image_size=(32, 48))
    assert len(train) == 3 and len(val) == 1
    item = train[0]
    assert item["img1"].shape == (3, 32, 48)
    assert item["flow"].shape == (2, 32, 48)
    assert 0 <= item["img1"].min() and item["img1"].max() <= 255


def test_sintel_loader_pair_and_volume(tmp_path):
    from deepof_amd.data import SintelDataset

    for scene in ("alley_1", "bamboo_2"):
        sdir = tmp_path / "training" / "clean" / scene
        fdir = tmp_path / "training" / "flow" / scene
        sdir.mkdir(parents=True)
        fdir.mkdir(parents=True)
        for i in range(1, 5):
            _save_img(sdir / f"frame_{i:04d}.png")
            write_flo(fdir / f"frame_{i:04d}.flo",
                      np.random.randn(32, 48, 2).astype(np.float32))

    pair = SintelDataset(str(tmp_path), "train", "clean", 2,
                         image_size=(32, 48))
    assert len(pair) == 6  # 3 windows per scene
    item = pair[0]
    assert item["img1"].shape == (3, 32, 48)
    assert item["flow"].shape == (2, 32, 48)

    vol = SintelDataset(str(tmp_path), "train", "clean", 3,
                        image_size=(32, 48))
    item = vol[0]
    assert item["volume"].shape == (9, 32, 48)
    assert item["flow"].shape == (4, 32, 48)  # 2 pairs

    val = SintelDataset(str(tmp_path), "val", "clean", 2,
                        image_size=(32, 48))
    assert len(val) == 2  # one window per scene fallback


def test_ucf101_loader(tmp_path):
    from deepof_amd.data import UCF101Dataset

    for cls in ("ApplyEyeMakeup", "Archery"):
        for g in (5, 9):  # group 5 -> test, 9 -> train
            clip = tmp_path / "frames" / cls / f"v_{cls}_g{g:02d}_c01"
            clip.mkdir(parents=True)
            for i in range(3):
                _save_img(clip / f"frame{i:03d}.jpg")

    train = UCF101Dataset(str(tmp_path), "train", image_size=(32, 48))
    test = UCF101Dataset(str(tmp_path), "test", image_size=(32, 48))
    assert len(train.clips) == 2 and len(test.clips) == 2
    item = train[0]
    assert item["img1"].shape == (3, 32, 48)
    assert item["label"].dtype == torch.long
    assert item["label"] in (0, 1)
