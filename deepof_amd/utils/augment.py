"""Batched GPU augmentation.

Parity: the reference applies geometric then photometric augmentation
in-graph per sample (/root/reference/version1/utils/augmentation.py:3-105
— crop-translate +-0.2, scale 0.9-2.0, LR flip; contrast +-0.3,
brightness sigma 0.2, per-channel color 0.9-1.1, gamma 0.7-1.5, additive
noise sigma<=0.04) and, in the v0 VGG config, feeds the photometrically
augmented pair to the network while the loss warps the geometry-only
pair (/root/reference/flyingChairsTrain_vgg.py:105-111).

Here the whole batch is transformed at once on the GPU: one
affine_grid/grid_sample pair for the geometric part (identical
transform for both frames of a pair), vectorized elementwise ops for
the photometric part.  Operates on RAW 0-255 images; returns
(geo1, geo2, photo1, photo2).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def geometric_augment(img1, img2, translate=0.2, scale_range=(0.9, 2.0),
                      flip_prob=0.5, generator=None):
    """Identical random affine (translate+scale) + LR flip per pair."""
    b, c, h, w = img1.shape
    dev = img1.device
    tx = (torch.rand(b, device=dev, generator=generator) * 2 - 1) * translate
    ty = (torch.rand(b, device=dev, generator=generator) * 2 - 1) * translate
    s = scale_range[0] + torch.rand(b, device=dev, generator=generator) * (
        scale_range[1] - scale_range[0])
    flip = torch.rand(b, device=dev, generator=generator) < flip_prob
    sign = torch.where(flip, -torch.ones_like(s), torch.ones_like(s))

    # output->input mapping: x_in = (x_out * sign) / s + tx
    theta = torch.zeros(b, 2, 3, device=dev, dtype=torch.float32)
    theta[:, 0, 0] = sign / s
    theta[:, 1, 1] = 1.0 / s
    theta[:, 0, 2] = tx
    theta[:, 1, 2] = ty
    grid = F.affine_grid(theta, [b, c, h, w], align_corners=False)

    def warp(img):
        return F.grid_sample(img.float(), grid, mode="bilinear",
                             padding_mode="border", align_corners=False)

    return warp(img1), warp(img2)


def photometric_augment(img1, img2, contrast=0.3, brightness_sigma=0.2,
                        color_range=(0.9, 1.1), gamma_range=(0.7, 1.5),
                        noise_sigma=0.04, generator=None):
    """Shared photometric transform per pair, on 0-255 images."""
    b = img1.shape[0]
    dev = img1.device
    co = 1.0 + (torch.rand(b, 1, 1, 1, device=dev, generator=generator) * 2
                - 1) * contrast
    br = torch.randn(b, 1, 1, 1, device=dev,
                     generator=generator) * brightness_sigma * 255.0
    col = color_range[0] + torch.rand(b, 3, 1, 1, device=dev,
                                      generator=generator) * (
        color_range[1] - color_range[0])
    gam = gamma_range[0] + torch.rand(b, 1, 1, 1, device=dev,
                                      generator=generator) * (
        gamma_range[1] - gamma_range[0])

    def apply(img):
        x = img.float() / 255.0
        mean = x.mean(dim=(2, 3), keepdim=True)
        x = (x - mean) * co + mean + br / 255.0
        x = x * col
        x = torch.clamp(x, 0.0, 1.0).pow(gam)
        if noise_sigma > 0:
            x = x + torch.randn(x.shape, device=dev,
                                generator=generator) * noise_sigma
        return torch.clamp(x, 0.0, 1.0) * 255.0

    return apply(img1), apply(img2)


def augment_pair(img1, img2, generator=None):
    """Full reference pipeline: geo (shared) then photo on a copy.

    Returns (geo1, geo2, photo1, photo2): the loss warps the geo pair,
    the network sees the photo pair (Flownet.py:39-40 /
    flyingChairsTrain_vgg.py:105-111 semantics).
    """
    geo1, geo2 = geometric_augment(img1, img2, generator=generator)
    photo1, photo2 = photometric_augment(geo1, geo2, generator=generator)
    return geo1, geo2, photo1, photo2
