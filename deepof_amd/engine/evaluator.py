"""Evaluation: the reference AEE protocol.

Parity (/root/reference/flyingChairsTrain.py:264-296): the final flow is
pr1 (finest prediction, at HALF input resolution) times its flow scale,
times 2 ("pr1 is still half of the final predicted flow value"), clipped
to the dataset's training flow range, bilinearly resized to the
ground-truth resolution WITHOUT magnitude rescaling (a reference quirk,
preserved for comparability), then AEE = mean endpoint error.
"""

from __future__ import annotations

import torch

from .. import ops
from ..losses.unsup import preprocess_images

# dataset -> (multiplier, clip_min, clip_max) — SURVEY §2.5 eval row
EVAL_POSTPROC = {
    "flying_chairs": (2.0, -300.0, 250.0),
    "synthetic": (2.0, -300.0, 250.0),
    "sintel": (3.0, -420.621, 426.311),
    "ucf101": (2.0, -300.0, 250.0),
}


def _postproc(dataset, mult=None, clip=None):
    """Dataset defaults with optional per-config overrides (the
    reference varies mult/clip per model variant)."""
    d_mult, d_min, d_max = EVAL_POSTPROC.get(
        dataset, EVAL_POSTPROC["flying_chairs"])
    if mult is not None:
        d_mult = mult
    if clip is not None:
        d_min, d_max = clip
    return d_mult, d_min, d_max


@torch.no_grad()
def predict_flow(model, img1_raw, img2_raw, mean_bgr, flow_scale_finest,
                 dataset: str = "flying_chairs",
                 gt_size: tuple[int, int] | None = None,
                 mult: float | None = None,
                 clip: tuple | None = None) -> torch.Tensor:
    """Run the model and apply the eval post-processing; returns [B,2,H,W]."""
    x1 = preprocess_images(img1_raw.float(), mean_bgr)
    x2 = preprocess_images(img2_raw.float(), mean_bgr)
    x = torch.cat([x1, x2], dim=1)
    if x.is_cuda:  # NHWC so eval hits the MFMA kernels, not a fallback
        x = x.contiguous(memory_format=torch.channels_last)
    out = model(x)
    flows = out[0] if isinstance(out, tuple) else out  # joint models
    mult, cmin, cmax = _postproc(dataset, mult, clip)
    pred = flows[0].float() * flow_scale_finest * mult
    pred = pred.clamp(cmin, cmax)
    if gt_size is not None and tuple(pred.shape[-2:]) != tuple(gt_size):
        # bilinear resize with NO magnitude rescale (reference quirk)
        pred = ops.resize_bilinear(pred, gt_size[0], gt_size[1])
    return pred


@torch.no_grad()
def evaluate_aee(model, loader, mean_bgr, flow_scale_finest, device,
                 dataset: str = "flying_chairs", max_batches=None,
                 dump_dir: str | None = None, dump_every: int = 10,
                 mult: float | None = None,
                 clip: tuple | None = None) -> float:
    """AEE over a loader; optionally dump flow color maps, warped
    frames and predicted .flo files (parity with the reference's eval
    artifacts, flyingChairsTrain.py:272-291)."""
    model.eval()
    total = 0.0
    count = 0
    for i, batch in enumerate(loader):
        if max_batches is not None and i >= max_batches:
            break
        gt = batch["flow"].to(device, non_blocking=True).float()
        if "volume" in batch:  # Sintel multi-frame: score the first pair
            vol = batch["volume"].to(device, non_blocking=True)
            T = vol.shape[1] // 3
            img1 = vol[:, :3]
            img2 = vol[:, 3:6]
            mean = torch.as_tensor(mean_bgr, device=vol.device)
            x = (vol.float() - mean.repeat(T).view(1, -1, 1, 1)) / 255.0
            out = model(x)
            flows = out[0] if isinstance(out, tuple) else out
            m_, cmin, cmax = _postproc(dataset, mult, clip)
            pred = (flows[0][:, :2].float() * flow_scale_finest * m_
                    ).clamp(cmin, cmax)
            gt = gt[:, :2]
            if tuple(pred.shape[-2:]) != tuple(gt.shape[-2:]):
                pred = ops.resize_bilinear(pred, gt.shape[-2], gt.shape[-1])
        else:
            img1 = batch["img1"].to(device, non_blocking=True)
            img2 = batch["img2"].to(device, non_blocking=True)
            pred = predict_flow(model, img1, img2, mean_bgr,
                                flow_scale_finest, dataset,
                                gt_size=tuple(gt.shape[-2:]),
                                mult=mult, clip=clip)
        total += float(ops.endpoint_error_sum(pred, gt))
        count += gt.shape[0] * gt.shape[-2] * gt.shape[-1]
        if dump_dir is not None and i % dump_every == 0:
            _dump_artifacts(dump_dir, i, img1, img2, pred, gt)
    model.train()
    return total / max(count, 1)


def _dump_artifacts(dump_dir, batch_idx, img1, img2, pred, gt):
    import os

    import numpy as np
    from PIL import Image

    from ..utils import flow_to_color, write_flo

    os.makedirs(dump_dir, exist_ok=True)
    p = pred[0].permute(1, 2, 0).cpu().numpy()
    g = gt[0].permute(1, 2, 0).cpu().numpy()
    write_flo(os.path.join(dump_dir, f"b{batch_idx:04d}_pred.flo"), p)
    Image.fromarray(flow_to_color(p)).save(
        os.path.join(dump_dir, f"b{batch_idx:04d}_pred.jpg"))
    Image.fromarray(flow_to_color(g)).save(
        os.path.join(dump_dir, f"b{batch_idx:04d}_gt.jpg"))
    # warped frame 2 (reconstruction of frame 1) at image resolution
    flow_img = ops.resize_bilinear(pred[:1], img1.shape[-2], img1.shape[-1])
    recon = ops.warp_bilinear(img2[:1].float(), flow_img)
    rec = recon[0].permute(1, 2, 0).cpu().numpy()[:, :, ::-1]  # BGR->RGB
    Image.fromarray(np.clip(rec, 0, 255).astype(np.uint8)).save(
        os.path.join(dump_dir, f"b{batch_idx:04d}_warped.jpg"))


@torch.no_grad()
def evaluate_accuracy(model, loader, mean_bgr, device,
                      max_batches=None) -> float:
    """UCF101 action accuracy (parity: ucf101train.py:183, :273)."""
    from ..losses.unsup import preprocess_images

    model.eval()
    correct = total = 0
    for i, batch in enumerate(loader):
        if max_batches is not None and i >= max_batches:
            break
        img1 = batch["img1"].to(device, non_blocking=True)
        img2 = batch["img2"].to(device, non_blocking=True)
        labels = batch["label"].to(device, non_blocking=True)
        x = torch.cat([preprocess_images(img1.float(), mean_bgr),
                       preprocess_images(img2.float(), mean_bgr)], dim=1)
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
        out = model(x)
        logits = out[1] if isinstance(out, tuple) else out
        correct += int((logits.argmax(1) == labels).sum())
        total += labels.numel()
    model.train()
    return correct / max(total, 1)
