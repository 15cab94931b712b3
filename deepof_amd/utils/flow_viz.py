"""Middlebury flow -> color visualization.

Vectorized re-implementation of the reference's colorwheel pipeline
(/root/reference/utils.py:209-350): 55-entry RY/YG/GC/CB/BM/MR wheel,
radius-normalized saturation, angle-indexed hue.  Returns uint8 RGB.
"""

from __future__ import annotations

import numpy as np

UNKNOWN_FLOW_THRESH = 1e9


def make_color_wheel() -> np.ndarray:
    """The 55-entry Middlebury color wheel, [55, 3] uint8-range floats."""
    RY, YG, GC, CB, BM, MR = 15, 6, 4, 11, 13, 6
    ncols = RY + YG + GC + CB + BM + MR
    wheel = np.zeros((ncols, 3), dtype=np.float64)
    col = 0
    # RY
    wheel[col : col + RY, 0] = 255
    wheel[col : col + RY, 1] = np.floor(255 * np.arange(RY) / RY)
    col += RY
    # YG
    wheel[col : col + YG, 0] = 255 - np.floor(255 * np.arange(YG) / YG)
    wheel[col : col + YG, 1] = 255
    col += YG
    # GC
    wheel[col : col + GC, 1] = 255
    wheel[col : col + GC, 2] = np.floor(255 * np.arange(GC) / GC)
    col += GC
    # CB
    wheel[col : col + CB, 1] = 255 - np.floor(255 * np.arange(CB) / CB)
    wheel[col : col + CB, 2] = 255
    col += CB
    # BM
    wheel[col : col + BM, 2] = 255
    wheel[col : col + BM, 0] = np.floor(255 * np.arange(BM) / BM)
    col += BM
    # MR
    wheel[col : col + MR, 2] = 255 - np.floor(255 * np.arange(MR) / MR)
    wheel[col : col + MR, 0] = 255
    return wheel


_WHEEL = make_color_wheel()


def compute_color(u: np.ndarray, v: np.ndarray) -> np.ndarray:
    """Color-code a normalized (|f| <= 1) flow field. Returns uint8 RGB."""
    nan_idx = np.isnan(u) | np.isnan(v)
    u = np.where(nan_idx, 0.0, u)
    v = np.where(nan_idx, 0.0, v)

    ncols = _WHEEL.shape[0]
    rad = np.sqrt(u**2 + v**2)
    a = np.arctan2(-v, -u) / np.pi  # [-1, 1]
    fk = (a + 1.0) / 2.0 * (ncols - 1)  # [0, ncols-1]
    k0 = np.floor(fk).astype(np.int64)
    k1 = (k0 + 1) % ncols
    f = fk - k0

    img = np.zeros(u.shape + (3,), dtype=np.uint8)
    for c in range(3):
        col0 = _WHEEL[k0, c] / 255.0
        col1 = _WHEEL[k1, c] / 255.0
        col = (1.0 - f) * col0 + f * col1
        inside = rad <= 1.0
        col = np.where(inside, 1.0 - rad * (1.0 - col), col * 0.75)
        col = np.where(nan_idx, 0.0, col)
        img[..., c] = np.floor(255.0 * col).astype(np.uint8)
    return img


def flow_to_color(flow: np.ndarray, max_rad: float | None = None) -> np.ndarray:
    """[H, W, 2] flow -> uint8 RGB image, normalized by the max radius."""
    flow = np.asarray(flow, dtype=np.float64)
    u = flow[..., 0].copy()
    v = flow[..., 1].copy()
    unknown = (np.abs(u) > UNKNOWN_FLOW_THRESH) | (np.abs(v) > UNKNOWN_FLOW_THRESH)
    u[unknown] = 0.0
    v[unknown] = 0.0
    rad = np.sqrt(u**2 + v**2)
    maxrad = max_rad if max_rad is not None else rad.max() if rad.size else 1.0
    eps = np.finfo(np.float64).eps
    img = compute_color(u / (maxrad + eps), v / (maxrad + eps))
    img[unknown] = 0
    return img
