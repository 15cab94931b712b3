"""Pure-PyTorch reference implementations of every deepof_amd hot op.

These are the numerics anchors: each HIP kernel in deepof_amd.ops.hip is
unit-tested against these fp32 implementations.  Semantics reproduce the
reference TF graph fragments (cited per function), vectorized (the
reference builds a B*C-unrolled gather graph, its worst inefficiency —
/root/reference/flyingChairsWrapFlow.py:800-838).

Layout convention: NCHW.  Flow tensors are [B, 2, H, W] with channel 0 =
u (horizontal displacement, +x along width) and channel 1 = v (vertical,
+y along height) — matching the reference's (U, V) channel order
(/root/reference/version1/model/warpflow.py:55-56).
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# Local response normalization (TF across-channel semantics)
# ---------------------------------------------------------------------------
def lrn(
    x: torch.Tensor,
    depth_radius: int = 4,
    bias: float = 1.0,
    alpha: float = 1.0,
    beta: float = 0.7,
) -> torch.Tensor:
    """tf.nn.local_response_normalization on NCHW input.

    sqr_sum[c] = sum over channels [c-r, c+r] of x^2 (alpha NOT divided
    by window size, per TF); out = x / (bias + alpha * sqr_sum) ** beta.
    Reference use: /root/reference/flyingChairsWrapFlow.py:25-26
    (depth_radius=4, beta=0.7 on 3-channel images, so the window spans
    all channels at every c).
    """
    b, c, h, w = x.shape
    x2 = (x * x).reshape(b, 1, c, h * w)
    kernel = 2 * depth_radius + 1
    sqr_sum = F.conv2d(
        F.pad(x2, (0, 0, depth_radius, depth_radius)),
        torch.ones(1, 1, kernel, 1, dtype=x.dtype, device=x.device),
    ).reshape(b, c, h, w)
    return x / torch.pow(bias + alpha * sqr_sum, beta)


# ---------------------------------------------------------------------------
# Bilinear resize (legacy TF resize_bilinear semantics)
# ---------------------------------------------------------------------------
def resize_bilinear(x: torch.Tensor, out_h: int, out_w: int) -> torch.Tensor:
    """tf.image.resize_bilinear (TF 0.x, align_corners=False) on NCHW.

    Source coordinate = out_index * (in_size / out_size) — the legacy TF
    mapping (no half-pixel offset; differs from torch's
    F.interpolate(align_corners=False)).  Taps clamp to the edge.
    Reference use: per-scale image pyramid
    (/root/reference/flyingChairsWrapFlow.py:61-62 etc.).
    """
    b, c, in_h, in_w = x.shape
    if (in_h, in_w) == (out_h, out_w):
        return x
    dev = x.device
    scale_y = in_h / out_h
    scale_x = in_w / out_w
    src_y = torch.arange(out_h, device=dev, dtype=torch.float32) * scale_y
    src_x = torch.arange(out_w, device=dev, dtype=torch.float32) * scale_x
    y0 = src_y.floor().long().clamp_(0, in_h - 1)
    x0 = src_x.floor().long().clamp_(0, in_w - 1)
    y1 = (y0 + 1).clamp_(0, in_h - 1)
    x1 = (x0 + 1).clamp_(0, in_w - 1)
    fy = (src_y - y0.to(torch.float32)).to(x.dtype).view(1, 1, out_h, 1)
    fx = (src_x - x0.to(torch.float32)).to(x.dtype).view(1, 1, 1, out_w)

    top = x[:, :, y0][:, :, :, x0] * (1 - fx) + x[:, :, y0][:, :, :, x1] * fx
    bot = x[:, :, y1][:, :, :, x0] * (1 - fx) + x[:, :, y1][:, :, :, x1] * fx
    return top * (1 - fy) + bot * fy


# ---------------------------------------------------------------------------
# Border / smoothness masks
# ---------------------------------------------------------------------------
def border_mask(h: int, w: int, ratio: float = 0.1, device=None) -> torch.Tensor:
    """[H, W] mask: 1 inside, 0 in a border of ceil(h * ratio) pixels.

    The border width uses the HEIGHT for all four sides (the reference's
    `shortestDim = height` quirk, flyingChairsWrapFlow.py:765-771).
    """
    bw = math.ceil(h * ratio)
    m = torch.zeros(h, w, device=device)
    m[bw : h - bw, bw : w - bw] = 1.0
    return m


def smoothness_mask(h: int, w: int, device=None) -> torch.Tensor:
    """[2, H, W]: ch0 zeros the last column (horizontal diffs), ch1 the
    last row (vertical diffs).  flyingChairsWrapFlow.py:774-779."""
    m = torch.ones(2, h, w, device=device)
    m[0, :, w - 1] = 0.0
    m[1, h - 1, :] = 0.0
    return m


# ---------------------------------------------------------------------------
# Bilinear backward warp
# ---------------------------------------------------------------------------
def warp_bilinear(img2: torch.Tensor, flow: torch.Tensor) -> torch.Tensor:
    """Backward-warp img2 by flow: out(y, x) = img2(y + v, x + u), bilinear.

    Integer tap indices are clipped to the edge independently (so at the
    border the four taps may coincide), and the bilinear weights come
    from the fractional part of the flow — exactly the reference graph
    (/root/reference/version1/model/warpflow.py:60-89).  Gradients flow
    to both img2 (tap values) and flow (fractional weights).

    img2: [B, C, H, W]; flow: [B, 2, H, W] (u, v), already scaled to
    pixels at this resolution.
    """
    b, c, h, w = img2.shape
    dev = img2.device
    u = flow[:, 0]  # [B, H, W]
    v = flow[:, 1]
    gy, gx = torch.meshgrid(
        torch.arange(h, device=dev, dtype=flow.dtype),
        torch.arange(w, device=dev, dtype=flow.dtype),
        indexing="ij",
    )
    fx = gx + u  # absolute sample x
    fy = gy + v
    x0 = torch.floor(fx)
    y0 = torch.floor(fy)
    xw = fx - x0  # fractional weights (autograd reaches flow through these)
    yw = fy - y0

    x0i = x0.detach().long()
    y0i = y0.detach().long()
    x1i = (x0i + 1).clamp(0, w - 1)
    y1i = (y0i + 1).clamp(0, h - 1)
    x0i = x0i.clamp(0, w - 1)
    y0i = y0i.clamp(0, h - 1)

    flat = img2.reshape(b, c, h * w)
    idx_a = (y0i * w + x0i).reshape(b, 1, h * w).expand(b, c, h * w)
    idx_b = (y1i * w + x0i).reshape(b, 1, h * w).expand(b, c, h * w)
    idx_c = (y0i * w + x1i).reshape(b, 1, h * w).expand(b, c, h * w)
    idx_d = (y1i * w + x1i).reshape(b, 1, h * w).expand(b, c, h * w)
    Ia = torch.gather(flat, 2, idx_a).reshape(b, c, h, w)
    Ib = torch.gather(flat, 2, idx_b).reshape(b, c, h, w)
    Ic = torch.gather(flat, 2, idx_c).reshape(b, c, h, w)
    Id = torch.gather(flat, 2, idx_d).reshape(b, c, h, w)

    xw = xw.unsqueeze(1)
    yw = yw.unsqueeze(1)
    return (
        Ia * (1 - xw) * (1 - yw)
        + Ib * (1 - xw) * yw
        + Ic * xw * (1 - yw)
        + Id * xw * yw
    )


# ---------------------------------------------------------------------------
# Charbonnier photometric loss
# ---------------------------------------------------------------------------
def charbonnier_photometric(
    recon: torch.Tensor,
    img1: torch.Tensor,
    epsilon: float = 1e-4,
    alpha_c: float = 0.25,
    mask: torch.Tensor | None = None,
) -> tuple[torch.Tensor, torch.Tensor]:
    """((255*(recon - img1))^2 + eps^2)^alpha_c, border-masked mean.

    Returns (loss, num_valid) where num_valid counts mask==1 elements
    over B*H*W*C (the reference divides BOTH the photometric and the
    smoothness sums by counts derived from this — warpflow.py:126-130).
    """
    diff = 255.0 * (recon - img1)
    ew = torch.pow(diff * diff + epsilon * epsilon, alpha_c)
    if mask is not None:
        m = mask.unsqueeze(0).unsqueeze(0)  # [1,1,H,W]
        ew = ew * m
        num_valid = m.expand_as(ew).sum()
        return ew.sum() / num_valid, num_valid
    return ew.mean(), torch.tensor(float(ew.numel()))


# ---------------------------------------------------------------------------
# Smoothness loss (first-order flow gradients, Charbonnier penalty)
# ---------------------------------------------------------------------------
def smoothness_loss(
    flow: torch.Tensor,
    num_valid_flows: torch.Tensor | float,
    epsilon: float = 1e-4,
    alpha_s: float = 0.37,
    apply_border_mask: bool = True,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Forward-difference flow smoothness, version1 depthwise semantics.

    dx(y, x) = f(y, x) - f(y, x+1); dy(y, x) = f(y, x) - f(y+1, x)
    (center-minus-right / center-minus-below stencils,
    /root/reference/version1/model/Flownet.py:75-82).  Each penalized as
    (d^2 + eps^2)^alpha_s, masked (last column for dx, last row for dy,
    plus the 10% border mask on both — warpflow.py:139-158), summed and
    divided by num_valid_flows = num_valid_pixels / 3 * 2.

    Returns (U_loss, V_loss) for the u and v channels.
    """
    b, _, h, w = flow.shape
    # Forward differences; the zero-filled last column/row reproduce the
    # smoothnessMask-on-delta (applied BEFORE the pow in the reference).
    dx = torch.zeros_like(flow)
    dy = torch.zeros_like(flow)
    dx[:, :, :, : w - 1] = flow[:, :, :, : w - 1] - flow[:, :, :, 1:]
    dy[:, :, : h - 1, :] = flow[:, :, : h - 1, :] - flow[:, :, 1:, :]

    eps2 = epsilon * epsilon

    def _charb(d):
        return torch.pow(d * d + eps2, alpha_s)

    if apply_border_mask:
        # borderMaskFlow multiplies AFTER the pow (warpflow.py:139-158),
        # so masked-out positions drop their eps^(2*alpha_s) term too.
        # The zero-filled last row/col sit inside the border and die here.
        bm = border_mask(h, w, device=flow.device).to(flow.dtype)
        u_ew = (_charb(dx[:, 0]) + _charb(dy[:, 0])) * bm
        v_ew = (_charb(dx[:, 1]) + _charb(dy[:, 1])) * bm
        u_loss = u_ew.sum() / num_valid_flows
        v_loss = v_ew.sum() / num_valid_flows
    else:
        u_loss = (_charb(dx[:, 0]).mean() + _charb(dy[:, 0]).mean()) / 2
        v_loss = (_charb(dx[:, 1]).mean() + _charb(dy[:, 1]).mean()) / 2
    return u_loss, v_loss


# ---------------------------------------------------------------------------
# Correlation cost volume (FlowNetC)
# ---------------------------------------------------------------------------
def correlation(
    f1: torch.Tensor,
    f2: torch.Tensor,
    max_displacement: int = 10,
) -> torch.Tensor:
    """FlowNetC correlation: out[b, d, y, x] = <f1[b,:,y,x], f2[b,:,y+dy,x+dx]> / C
    for (dy, dx) in [-md, md]^2 (d = (dy+md)*(2md+1) + (dx+md)).

    Absent in the reference TF repo; required by BASELINE.json configs[2]
    (441-channel cost volume, md=10).  Zero padding outside f2.
    """
    b, c, h, w = f1.shape
    md = max_displacement
    k = 2 * md + 1
    f2p = F.pad(f2, (md, md, md, md))
    out = f1.new_zeros(b, k * k, h, w)
    for dy in range(-md, md + 1):
        for dx in range(-md, md + 1):
            d = (dy + md) * k + (dx + md)
            f2s = f2p[:, :, dy + md : dy + md + h, dx + md : dx + md + w]
            out[:, d] = (f1 * f2s).sum(dim=1) / c
    return out


# ---------------------------------------------------------------------------
# Full per-scale unsupervised loss (warp + photometric + smoothness)
# ---------------------------------------------------------------------------
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
    """One pyramid scale of the unsupervised loss.

    Reproduces loss_interp (/root/reference/version1/model/warpflow.py:4-173,
    v0 flyingChairsWrapFlow.py:752-876): scale the raw flow prediction,
    backward-warp img2, masked Charbonnier photometric, masked
    first-order smoothness on the SCALED flow (v0 semantics; version1
    uses the raw flow — we follow v0, the trained configuration).

    Returns dict(total, photo, u_loss, v_loss[, recon]).
    """
    b, c, h, w = img1.shape
    scaled = flow_raw * flow_scale
    recon = warp_bilinear(img2, scaled)
    bw = math.ceil(h * 0.1)
    masked = (h - 2 * bw) > 0 and (w - 2 * bw) > 0
    bm = border_mask(h, w, device=img1.device).to(img1.dtype) if masked else None
    photo, num_valid = charbonnier_photometric(recon, img1, epsilon, alpha_c, bm)
    num_valid_flows = num_valid / c * 2
    u_loss, v_loss = smoothness_loss(scaled, num_valid_flows, epsilon, alpha_s,
                                     apply_border_mask=masked)
    total = photo + lambda_smooth * (u_loss + v_loss)
    out = {"total": total, "photo": photo, "u_loss": u_loss, "v_loss": v_loss}
    if return_recon:
        out["recon"] = recon
    return out


# ---------------------------------------------------------------------------
# Edge-aware smoothness weighting (loss_interp_bk / needImageGradients)
# ---------------------------------------------------------------------------
def image_gradient_masks(img1: torch.Tensor) -> torch.Tensor:
    """exp-style edge weights (1 - |normalized Sobel gradient|).

    Reproduces the needImageGradients branch
    (/root/reference/version1/model/warpflow.py:92-117): per-image
    0-255 rescale, grayscale, Sobel x/y, normalize by the max |g|,
    weight = 1 - |g|.  Returns [B, 2, H, W] (x-weight, y-weight).
    """
    b = img1.shape[0]
    mn = img1.amin(dim=(1, 2, 3), keepdim=True)
    mx = img1.amax(dim=(1, 2, 3), keepdim=True)
    x = (img1 - mn) / (mx - mn + 1e-12) * 255.0
    gray = x.mean(dim=1, keepdim=True)
    sob_x = torch.tensor([[-1.0, 0, 1], [-2, 0, 2], [-1, 0, 1]],
                         device=img1.device).view(1, 1, 3, 3)
    sob_y = sob_x.transpose(2, 3)
    gx = F.conv2d(gray, sob_x, padding=1)
    gy = F.conv2d(gray, sob_y, padding=1)
    gx = gx / (gx.abs().amax(dim=(1, 2, 3), keepdim=True) + 1e-12)
    gy = gy / (gy.abs().amax(dim=(1, 2, 3), keepdim=True) + 1e-12)
    return torch.cat([1.0 - gx.abs(), 1.0 - gy.abs()], dim=1)


def unsup_loss_scale_edge_aware(
    flow_raw, img1, img2, flow_scale, epsilon=1e-4, alpha_c=0.25,
    alpha_s=0.37, lambda_smooth=1.0,
):
    """unsup_loss_scale with edge-aware smoothness: the per-direction
    Charbonnier terms are multiplied by the (1-|gradient|) masks before
    the border mask (warpflow.py:147-158)."""
    b, c, h, w = img1.shape
    scaled = flow_raw * flow_scale
    recon = warp_bilinear(img2, scaled)
    bw = math.ceil(h * 0.1)
    masked = (h - 2 * bw) > 0 and (w - 2 * bw) > 0
    bm = border_mask(h, w, device=img1.device).to(img1.dtype) if masked else None
    photo, num_valid = charbonnier_photometric(recon, img1, epsilon, alpha_c, bm)
    num_valid_flows = num_valid / c * 2

    dx = torch.zeros_like(scaled)
    dy = torch.zeros_like(scaled)
    dx[:, :, :, : w - 1] = scaled[:, :, :, : w - 1] - scaled[:, :, :, 1:]
    dy[:, :, : h - 1, :] = scaled[:, :, : h - 1, :] - scaled[:, :, 1:, :]
    eps2 = epsilon * epsilon
    gmask = image_gradient_masks(img1)
    wx = gmask[:, 0:1]
    wy = gmask[:, 1:2]
    ew_u = torch.pow(dx[:, 0:1] ** 2 + eps2, alpha_s) * wx + \
        torch.pow(dy[:, 0:1] ** 2 + eps2, alpha_s) * wy
    ew_v = torch.pow(dx[:, 1:2] ** 2 + eps2, alpha_s) * wx + \
        torch.pow(dy[:, 1:2] ** 2 + eps2, alpha_s) * wy
    if masked:
        u_loss = (ew_u[:, 0] * bm).sum() / num_valid_flows
        v_loss = (ew_v[:, 0] * bm).sum() / num_valid_flows
    else:
        u_loss = ew_u.mean()
        v_loss = ew_v.mean()
    total = photo + lambda_smooth * (u_loss + v_loss)
    return {"total": total, "photo": photo, "u_loss": u_loss,
            "v_loss": v_loss}
