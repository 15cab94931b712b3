from .functional import (
    warp_bilinear,
    resize_bilinear,
    lrn,
    correlation,
    unsup_loss_scale,
    endpoint_error_sum,
    hip_available,
    require_hip,
)

__all__ = [
    "warp_bilinear",
    "resize_bilinear",
    "lrn",
    "correlation",
    "unsup_loss_scale",
    "endpoint_error_sum",
    "hip_available",
    "require_hip",
]
