"""Flow error metrics.

Semantics match the reference (/root/reference/utils.py:64-80):
- endpoint_error (AEE): mean over all pixels of sqrt(du^2 + dv^2).
- angular_error: mean arccos of the normalized (u, v, 1) inner product.

Both accept torch tensors (any device) or numpy arrays, with the flow
channel either last ([..., H, W, 2]) or as dim 1 ([B, 2, H, W]).
"""

from __future__ import annotations

import numpy as np
import torch


def _as_uv(flow):
    """Return (u, v) tensors from [..., 2]-last or [B, 2, H, W] layouts."""
    if isinstance(flow, np.ndarray):
        flow = torch.from_numpy(flow)
    if flow.shape[-1] == 2:
        return flow[..., 0], flow[..., 1]
    if flow.dim() == 4 and flow.shape[1] == 2:
        return flow[:, 0], flow[:, 1]
    raise ValueError(f"cannot infer flow layout from shape {tuple(flow.shape)}")


def endpoint_error(flow, flow_gt) -> float:
    """Average endpoint error between two flow fields."""
    u1, v1 = _as_uv(flow)
    u2, v2 = _as_uv(flow_gt)
    u1, v1 = u1.float(), v1.float()
    u2, v2 = u2.float(), v2.float()
    ee = torch.sqrt((u1 - u2) ** 2 + (v1 - v2) ** 2)
    return float(ee.mean())


def angular_error(flow, flow_gt) -> float:
    """Average angular error (radians) between two flow fields."""
    u, v = _as_uv(flow)
    ug, vg = _as_uv(flow_gt)
    u, v, ug, vg = u.float(), v.float(), ug.float(), vg.float()
    num = 1.0 + u * ug + v * vg
    den = torch.sqrt(1.0 + u**2 + v**2) * torch.sqrt(1.0 + ug**2 + vg**2)
    ae = torch.arccos(torch.clamp(num / den, -1.0, 1.0))
    return float(ae.mean())
