"""Device-dispatching functional ops.

On GPU (torch "cuda" == ROCm/HIP here) every op runs the hand-written
CDNA4 HIP kernel from deepof_amd.ops.hip — and FAILS LOUDLY if the
extension is not built (no silent eager fallback on a GPU box).  On CPU
the pure-torch reference implementations run instead, so the same model
code is testable without a GPU.
"""

from __future__ import annotations

import torch

from . import reference as ref

_HIP = None
_HIP_ERR: str | None = None


def _load_hip():
    global _HIP, _HIP_ERR
    if _HIP is not None or _HIP_ERR is not None:
        return _HIP
    try:
        from .hip import _deepof_hip  # built in-tree by setup.py / __graft_entry__

        _HIP = _deepof_hip
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes only
        _HIP_ERR = str(e)
    return _HIP


def hip_available() -> bool:
    return _load_hip() is not None


def require_hip():
    m = _load_hip()
    if m is None:
        raise RuntimeError(
            "deepof_amd HIP extension is not built but a GPU tensor reached a "
            "deepof_amd op. Build it in-tree first: `python setup.py "
            f"build_ext --inplace` (import error: {_HIP_ERR})"
        )
    return m


def _on_gpu(*tensors) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---------------------------------------------------------------------------
# warp
# ---------------------------------------------------------------------------
class _WarpBilinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, img2, flow):
        hip = require_hip()
        recon = hip.warp_forward(img2.contiguous(), flow.contiguous())
        ctx.save_for_backward(img2, flow)
        return recon

    @staticmethod
    def backward(ctx, grad_out):
        img2, flow = ctx.saved_tensors
        hip = require_hip()
        gimg2, gflow = hip.warp_backward(
            grad_out.contiguous(), img2.contiguous(), flow.contiguous()
        )
        return gimg2, gflow


def warp_bilinear(img2: torch.Tensor, flow: torch.Tensor) -> torch.Tensor:
    """Backward-warp img2 ([B,C,H,W]) by pixel-scaled flow ([B,2,H,W])."""
    if _on_gpu(img2, flow):
        return _WarpBilinear.apply(img2, flow.float())
    return ref.warp_bilinear(img2, flow)


# ---------------------------------------------------------------------------
# resize / lrn (loss-branch image preprocessing; inputs are constants)
# ---------------------------------------------------------------------------
def resize_bilinear(x: torch.Tensor, out_h: int, out_w: int) -> torch.Tensor:
    if x.is_cuda:
        if x.requires_grad and torch.is_grad_enabled():
            # differentiable path (same legacy-TF sampling, torch ops);
            # the HIP kernel serves the no-grad image-pyramid hot path
            return ref.resize_bilinear(x, out_h, out_w)
        return require_hip().resize_bilinear(x.contiguous(), out_h, out_w)
    return ref.resize_bilinear(x, out_h, out_w)


def lrn(x, depth_radius: int = 4, bias: float = 1.0, alpha: float = 1.0,
        beta: float = 0.7) -> torch.Tensor:
    if x.is_cuda:
        if x.requires_grad and torch.is_grad_enabled():
            # differentiable torch-op path (a future loss backprops
            # through the image branch); the HIP kernel serves the
            # no-grad image-pyramid hot path
            return ref.lrn(x, depth_radius, bias, alpha, beta)
        return require_hip().lrn_forward(x.contiguous(), depth_radius, bias, alpha, beta)
    return ref.lrn(x, depth_radius, bias, alpha, beta)


# ---------------------------------------------------------------------------
# correlation (FlowNetC)
# ---------------------------------------------------------------------------
class _Correlation(torch.autograd.Function):
    @staticmethod
    def forward(ctx, f1, f2, md):
        hip = require_hip()
        out = hip.correlation_forward(f1.contiguous(), f2.contiguous(), md)
        ctx.save_for_backward(f1, f2)
        ctx.md = md
        return out

    @staticmethod
    def backward(ctx, grad_out):
        f1, f2 = ctx.saved_tensors
        hip = require_hip()
        gf1, gf2 = hip.correlation_backward(
            grad_out.contiguous(), f1.contiguous(), f2.contiguous(), ctx.md
        )
        return gf1, gf2, None


def correlation(f1: torch.Tensor, f2: torch.Tensor, max_displacement: int = 10):
    """FlowNetC cost volume, (2*md+1)^2 channels, zero-padded, /C."""
    if _on_gpu(f1, f2):
        return _Correlation.apply(f1, f2, max_displacement)
    return ref.correlation(f1, f2, max_displacement)


# ---------------------------------------------------------------------------
# fused per-scale unsupervised loss
# ---------------------------------------------------------------------------
class _FusedUnsupLoss(torch.autograd.Function):
    """HIP fused warp + Charbonnier photometric + smoothness for one scale.

    Forward returns raw SUMS (photo_sum, u_sum, v_sum) and the recon
    image; normalization and the lambda-weighted total happen in Python.
    Backward produces d(flow_raw) only — images are data, not params.
    """

    @staticmethod
    def forward(ctx, flow_raw, img1, img2, flow_scale, eps, alpha_c, alpha_s,
                want_recon):
        hip = require_hip()
        photo_sum, u_sum, v_sum, recon = hip.unsup_loss_forward(
            flow_raw.contiguous(), img1.contiguous(), img2.contiguous(),
            flow_scale, eps, alpha_c, alpha_s, want_recon,
        )
        ctx.save_for_backward(flow_raw, img1, img2)
        ctx.params = (flow_scale, eps, alpha_c, alpha_s)
        return photo_sum, u_sum, v_sum, recon

    @staticmethod
    def backward(ctx, g_photo, g_u, g_v, g_recon):
        flow_raw, img1, img2 = ctx.saved_tensors
        flow_scale, eps, alpha_c, alpha_s = ctx.params
        hip = require_hip()
        # upstream grads stay on device (no .item() sync; hipGraph-safe)
        dev = flow_raw.device
        def _g(t):
            return (t.float().reshape(1) if isinstance(t, torch.Tensor)
                    else torch.zeros(1, device=dev))
        gs = torch.cat([_g(g_photo), _g(g_u), _g(g_v)])
        gflow = hip.unsup_loss_backward(
            flow_raw.contiguous(), img1.contiguous(), img2.contiguous(),
            flow_scale, eps, alpha_c, alpha_s, gs,
        )
        return gflow, None, None, None, None, None, None, None


def unsup_loss_scale(
    flow_raw: torch.Tensor,
    img1: torch.Tensor,
    img2: torch.Tensor,
    flow_scale: float,
    epsilon: float = 1e-4,
    alpha_c: float = 0.25,
    alpha_s: float = 0.37,
    lambda_smooth: float = 1.0,
    return_recon: bool = False,
):
    """One pyramid scale of the unsupervised warp loss (see
    deepof_amd.ops.reference.unsup_loss_scale for the semantics spec)."""
    if not _on_gpu(flow_raw, img1, img2):
        return ref.unsup_loss_scale(
            flow_raw, img1, img2, flow_scale, epsilon, alpha_c, alpha_s,
            lambda_smooth, return_recon,
        )
    b, c, h, w = img1.shape
    import math

    # mirror the kernel's border-mask fallback for degenerate scales
    bw = math.ceil(h * 0.1)
    if (h - 2 * bw) > 0 and (w - 2 * bw) > 0:
        num_valid = float(b * c * (h - 2 * bw) * (w - 2 * bw))
        num_valid_flows = num_valid / c * 2
    else:
        num_valid = float(b * c * h * w)
        num_valid_flows = float(2 * b * h * w)
    photo_sum, u_sum, v_sum, recon = _FusedUnsupLoss.apply(
        flow_raw.float(), img1, img2, flow_scale, epsilon, alpha_c, alpha_s,
        return_recon,
    )
    photo = photo_sum / num_valid
    u_loss = u_sum / num_valid_flows
    v_loss = v_sum / num_valid_flows
    total = photo + lambda_smooth * (u_loss + v_loss)
    out = {"total": total, "photo": photo, "u_loss": u_loss, "v_loss": v_loss}
    if return_recon:
        out["recon"] = recon
    return out


# ---------------------------------------------------------------------------
# EPE reduction (eval)
# ---------------------------------------------------------------------------
def endpoint_error_sum(flow: torch.Tensor, flow_gt: torch.Tensor) -> torch.Tensor:
    """Sum over all pixels of sqrt(du^2+dv^2); divide by B*H*W for AEE."""
    if _on_gpu(flow, flow_gt):
        return require_hip().epe_sum(flow.contiguous().float(),
                                     flow_gt.contiguous().float())
    d = (flow.float() - flow_gt.float())
    return torch.sqrt(d[:, 0] ** 2 + d[:, 1] ** 2).sum()
