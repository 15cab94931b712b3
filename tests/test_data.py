import numpy as np
import torch

from deepof_amd.data import SyntheticFlowDataset, build_dataloader
from deepof_amd.data.synthetic import make_synthetic_pair
from deepof_amd.ops import reference as ref


def test_synthetic_shapes():
    ds = SyntheticFlowDataset(8, 32, 48)
    item = ds[0]
    assert item["img1"].shape == (3, 32, 48)
    assert item["img2"].shape == (3, 32, 48)
    assert item["flow"].shape == (2, 32, 48)
    assert 0 <= item["img1"].min() and item["img1"].max() <= 255


def test_synthetic_deterministic():
    ds = SyntheticFlowDataset(8, 16, 16, seed=3)
    a, b = ds[2], ds[2]
    torch.testing.assert_close(a["img1"], b["img1"])


def test_synthetic_flow_consistency():
    """Warping img2 back by the GT flow should reconstruct img1 far
    better than img2 itself does (the loss has real signal)."""
    rng = np.random.default_rng(0)
    img1, img2, flow = make_synthetic_pair(rng, 48, 64, max_flow=6.0)
    t1 = torch.from_numpy(img1).unsqueeze(0)
    t2 = torch.from_numpy(img2).unsqueeze(0)
    tf = torch.from_numpy(flow).unsqueeze(0)
    recon = ref.warp_bilinear(t2, tf)
    bm = ref.border_mask(48, 64)
    err_warp = ((recon - t1).abs() * bm).mean()
    err_base = ((t2 - t1).abs() * bm).mean()
    assert err_warp < 0.5 * err_base


def test_dataloader():
    ds = SyntheticFlowDataset(8, 16, 24)
    dl = build_dataloader(ds, 4, num_workers=0)
    batch = next(iter(dl))
    assert batch["img1"].shape == (4, 3, 16, 24)
