"""Multi-scale unsupervised photometric warp loss.

Reproduces the loss branch of the reference models
(/root/reference/flyingChairsWrapFlow.py:16-26 preprocessing,
:58-124 per-scale loss wiring): mean-subtract + /255, across-channel
LRN (depth_radius=4, beta=0.7), a bilinear image pyramid at every flow
prediction's resolution, then per scale the fused
warp + Charbonnier photometric + smoothness loss, weighted-summed with
the per-scale loss weights.

All image-branch math stays fp32 (the Charbonnier (x^2+eps^2)^0.25 with
eps=1e-4 underflows carelessly in bf16); flows are cast to fp32 at the
loss boundary.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops

# dataset BGR means
DATASET_MEANS = {
    "flying_chairs": (97.533268, 99.238236, 97.055973),
    "sintel": (70.1433, 83.1915, 92.8827),
    "ucf101": (104.0, 117.0, 123.0),
}


def preprocess_images(img: torch.Tensor, mean_bgr) -> torch.Tensor:
    """(img - mean) / 255 on a [B, 3, H, W] 0-255 BGR tensor."""
    mean = torch.as_tensor(mean_bgr, dtype=img.dtype, device=img.device)
    return (img - mean.view(1, 3, 1, 1)) / 255.0


class MultiScaleUnsupLoss(nn.Module):
    def __init__(
        self,
        flow_scales: list[float],
        loss_weights: list[float],
        mean_bgr=DATASET_MEANS["flying_chairs"],
        epsilon: float = 1e-4,
        alpha_c: float = 0.25,
        alpha_s: float = 0.37,
        lambda_smooth: float = 1.0,
        edge_aware: bool = False,
    ):
        super().__init__()
        assert len(flow_scales) == len(loss_weights)
        self.edge_aware = edge_aware
        self.flow_scales = flow_scales
        self.loss_weights = loss_weights
        self.mean_bgr = mean_bgr
        self.epsilon = epsilon
        self.alpha_c = alpha_c
        self.alpha_s = alpha_s
        self.lambda_smooth = lambda_smooth

    @torch.no_grad()
    def _pyramid(self, img1_raw, img2_raw, sizes):
        """Normalized + LRN'd image pair resized to each flow scale."""
        i1 = ops.lrn(preprocess_images(img1_raw.float(), self.mean_bgr))
        i2 = ops.lrn(preprocess_images(img2_raw.float(), self.mean_bgr))
        return (
            [ops.resize_bilinear(i1, h, w) for (h, w) in sizes],
            [ops.resize_bilinear(i2, h, w) for (h, w) in sizes],
        )

    def forward(self, flows: list[torch.Tensor], img1_raw: torch.Tensor,
                img2_raw: torch.Tensor, want_recon: bool = False):
        """flows: raw predictions FINEST FIRST; img*_raw: [B,3,H,W] 0-255.

        Returns dict with 'total' (the weighted scalar to backprop),
        'scales' (per-scale component dicts), 'flows_all' (scaled flow
        pyramid, finest first) and optionally 'recon'.
        """
        assert len(flows) == len(self.flow_scales)
        sizes = [tuple(f.shape[-2:]) for f in flows]
        pyr1, pyr2 = self._pyramid(img1_raw, img2_raw, sizes)

        total = None
        scale_losses = []
        recon = None
        for k, flow in enumerate(flows):
            if self.edge_aware:
                # loss_interp_bk variant: torch-op path (image-gradient
                # masks; the fused HIP kernel covers the default config)
                from ..ops.reference import unsup_loss_scale_edge_aware

                res = unsup_loss_scale_edge_aware(
                    flow.float(), pyr1[k], pyr2[k], self.flow_scales[k],
                    self.epsilon, self.alpha_c, self.alpha_s,
                    self.lambda_smooth,
                )
            else:
                res = ops.unsup_loss_scale(
                    flow.float(), pyr1[k], pyr2[k], self.flow_scales[k],
                    self.epsilon, self.alpha_c, self.alpha_s,
                    self.lambda_smooth,
                    return_recon=(want_recon and k == 0),
                )
            if want_recon and k == 0:
                recon = res.pop("recon")
            scale_losses.append(res)
            term = self.loss_weights[k] * res["total"]
            total = term if total is None else total + term

        out = {
            "total": total,
            "scales": scale_losses,
            "flows_all": [f.float() * s for f, s in zip(flows, self.flow_scales)],
        }
        if recon is not None:
            out["recon"] = recon
        return out


class MultiFrameUnsupLoss(nn.Module):
    """Multi-frame (Sintel volume) unsupervised loss.

    The reference's loss_interp_multi warps frame t+1 onto frame t for
    every consecutive pair inside one graph op
    (/root/reference/sintelWrapFlow.py:492-630, channel->flow index map
    :543, source channel c+3 :565).  Here each pair reuses the fused
    per-scale HIP loss on channel slices: flow pair t = channels
    [2t, 2t+1], images frame t = channels [3t : 3t+3].  Pair losses are
    averaged.  LRN is applied per frame (the reference LRN's
    depth_radius=4 window leaks across frame boundaries in the stacked
    volume — a quirk not replicated).
    """

    def __init__(self, flow_scales, loss_weights,
                 mean_bgr=DATASET_MEANS["sintel"], epsilon=1e-4,
                 alpha_c=0.3, alpha_s=0.3, lambda_smooth=0.0):
        super().__init__()
        self.flow_scales = flow_scales
        self.loss_weights = loss_weights
        self.mean_bgr = mean_bgr
        self.epsilon = epsilon
        self.alpha_c = alpha_c
        self.alpha_s = alpha_s
        self.lambda_smooth = lambda_smooth

    def forward(self, flows: list, volume_raw: torch.Tensor):
        """flows: raw [B, 2(T-1), h_k, w_k] finest first; volume_raw:
        [B, 3T, H, W] 0-255 BGR."""
        b, ct, H, W = volume_raw.shape
        T = ct // 3
        assert flows[0].shape[1] == 2 * (T - 1)
        mean = torch.as_tensor(self.mean_bgr, dtype=torch.float32,
                               device=volume_raw.device)
        vol = (volume_raw.float() - mean.repeat(T).view(1, ct, 1, 1)) / 255.0
        with torch.no_grad():
            frames = [ops.lrn(vol[:, 3 * t: 3 * t + 3].contiguous())
                      for t in range(T)]

        total = None
        scale_losses = []
        for k, flow in enumerate(flows):
            h, w = flow.shape[-2:]
            with torch.no_grad():
                pyr = [ops.resize_bilinear(f, h, w) for f in frames]
            pair_total = None
            for t in range(T - 1):
                res = ops.unsup_loss_scale(
                    flow[:, 2 * t: 2 * t + 2].float(), pyr[t], pyr[t + 1],
                    self.flow_scales[k], self.epsilon, self.alpha_c,
                    self.alpha_s, self.lambda_smooth,
                )
                pair_total = res["total"] if pair_total is None \
                    else pair_total + res["total"]
            loss_k = pair_total / (T - 1)
            scale_losses.append(loss_k)
            term = self.loss_weights[k] * loss_k
            total = term if total is None else total + term
        return {"total": total, "scales": scale_losses}
