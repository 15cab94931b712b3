"""Image loading helpers.

Images come back as float32 [H, W, 3] BGR 0-255 arrays — the reference
pipeline is cv2-based and all dataset means are BGR, so
the channel order is kept BGR end to end.
"""

from __future__ import annotations

import numpy as np
from PIL import Image


def load_image(path, size_hw: tuple[int, int] | None = None) -> np.ndarray:
    """Read an image file -> float32 [H, W, 3] BGR; optional bilinear
    resize to (H, W) (matching the loaders' cv2.resize,
    /root/reference/flyingChairsLoader.py:64-82)."""
    with Image.open(path) as im:
        im = im.convert("RGB")
        if size_hw is not None and (im.height, im.width) != size_hw:
            im = im.resize((size_hw[1], size_hw[0]), Image.BILINEAR)
        arr = np.asarray(im, dtype=np.float32)
    return arr[:, :, ::-1].copy()  # RGB -> BGR


def to_chw(img_hwc: np.ndarray) -> np.ndarray:
    return np.ascontiguousarray(img_hwc.transpose(2, 0, 1))
