"""Middlebury `.flo` optical-flow file I/O.

Format (kept byte-compatible with the reference implementation,
/root/reference/utils.py:4-52): little-endian float32 magic 202021.25,
then int32 width, int32 height, then h*w*2 float32 samples interleaved
as (u, v) in row-major order.  The reference's writer has an undefined
``TAG_CHAR`` bug (utils.py:44); this writer is complete.
"""

from __future__ import annotations

import os

import numpy as np

FLO_MAGIC = 202021.25


def read_flo(path: str | os.PathLike) -> np.ndarray:
    """Read a .flo file -> float32 array of shape [H, W, 2] (u, v)."""
    with open(path, "rb") as f:
        magic = np.fromfile(f, np.float32, count=1)
        if magic.size == 0 or magic[0] != np.float32(FLO_MAGIC):
            raise ValueError(f"{path}: bad .flo magic {magic!r} (want {FLO_MAGIC})")
        w = int(np.fromfile(f, np.int32, count=1)[0])
        h = int(np.fromfile(f, np.int32, count=1)[0])
        if not (0 < w < 100000 and 0 < h < 100000):
            raise ValueError(f"{path}: implausible .flo dims {w}x{h}")
        data = np.fromfile(f, np.float32, count=2 * w * h)
        if data.size != 2 * w * h:
            raise ValueError(f"{path}: truncated .flo (got {data.size} floats)")
    return data.reshape(h, w, 2)


def write_flo(path: str | os.PathLike, flow: np.ndarray) -> None:
    """Write a [H, W, 2] float32 flow array as a .flo file."""
    flow = np.asarray(flow)
    if flow.ndim != 3 or flow.shape[2] != 2:
        raise ValueError(f"flow must be [H, W, 2], got {flow.shape}")
    h, w = flow.shape[:2]
    with open(path, "wb") as f:
        np.float32(FLO_MAGIC).tofile(f)
        np.int32(w).tofile(f)
        np.int32(h).tofile(f)
        flow.astype(np.float32).tofile(f)
