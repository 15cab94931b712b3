"""Inference serving: a FastAPI app around a trained flow model.

    python -m deepof_amd serve --config cfg.yaml --checkpoint ckpt.pt \
        --host 0.0.0.0 --port 8000

Endpoints:
    GET  /healthz            -> {"status": "ok", "model": ..., "device": ...}
    POST /flow               JSON {"img1": <base64 PNG/JPEG>,
                                   "img2": <base64 PNG/JPEG>};
                             query: format=json|flo|png
        json (default): flow statistics + shape
        flo:            the predicted flow as Middlebury .flo bytes
        png:            Middlebury color-wheel visualization

The model runs the same eval path as the evaluator (reference
post-processing: pr1 x flow_scale x mult, clip —
/root/reference/flyingChairsTrain.py:264-266), in bf16 on GPU with
channels_last so the MFMA kernels serve the request.
"""

from __future__ import annotations

import io
import struct

import numpy as np
import torch
from pydantic import BaseModel


class FlowRequest(BaseModel):
    img1: str  # base64 PNG/JPEG
    img2: str


def _decode_image(data: bytes, size=None) -> torch.Tensor:
    """bytes -> [1, 3, H, W] float 0-255 BGR (the reference's cv2
    convention: loaders produce BGR)."""
    from PIL import Image

    img = Image.open(io.BytesIO(data)).convert("RGB")
    if size is not None:
        img = img.resize((size[1], size[0]), Image.BILINEAR)
    arr = np.asarray(img, dtype=np.float32)[:, :, ::-1]  # RGB -> BGR
    return torch.from_numpy(arr.copy()).permute(2, 0, 1).unsqueeze(0)


def _flo_bytes(flow_hw2: np.ndarray) -> bytes:
    """Middlebury .flo encoding (magic 202021.25, w, h, interleaved)."""
    h, w, _ = flow_hw2.shape
    out = io.BytesIO()
    out.write(struct.pack("<f", 202021.25))
    out.write(struct.pack("<ii", w, h))
    out.write(flow_hw2.astype("<f4").tobytes())
    return out.getvalue()


def create_app(model, flow_scales, mean_bgr, device,
               dataset: str = "flying_chairs", precision: str = "bf16",
               eval_mult=None, eval_clip=None):
    import base64
    import time

    from fastapi import Body, FastAPI
    from fastapi.responses import JSONResponse, Response

    from .engine.evaluator import predict_flow

    app = FastAPI(title="deepof_amd flow service")
    model.eval()
    use_bf16 = precision == "bf16" and device.type == "cuda"

    # Prometheus metrics (own registry so repeated create_app calls in
    # one process don't collide)
    from prometheus_client import (CONTENT_TYPE_LATEST, CollectorRegistry,
                                   Counter, Histogram, generate_latest)

    registry = CollectorRegistry()
    req_count = Counter("deepof_flow_requests_total",
                        "flow requests served", ["format"],
                        registry=registry)
    req_latency = Histogram("deepof_flow_latency_seconds",
                            "end-to-end /flow latency",
                            registry=registry)

    @app.get("/metrics")
    def metrics():
        return Response(content=generate_latest(registry),
                        media_type=CONTENT_TYPE_LATEST)

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "model": type(model).__name__,
                "device": str(device)}

    @app.post("/flow")
    def flow(req: FlowRequest = Body(...), format: str = "json"):
        t_start = time.perf_counter()
        req_count.labels(format=format).inc()
        t1 = _decode_image(base64.b64decode(req.img1)).to(device)
        t2 = _decode_image(base64.b64decode(req.img2),
                           size=t1.shape[-2:]).to(device)
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=use_bf16):
            pred = predict_flow(model, t1, t2, mean_bgr, flow_scales[0],
                                dataset, gt_size=tuple(t1.shape[-2:]),
                                mult=eval_mult, clip=eval_clip)
        f = pred[0].float().permute(1, 2, 0).cpu().numpy()
        req_latency.observe(time.perf_counter() - t_start)
        if format == "flo":
            return Response(content=_flo_bytes(f),
                            media_type="application/octet-stream")
        if format == "png":
            from PIL import Image

            from .utils import flow_to_color

            buf = io.BytesIO()
            Image.fromarray(flow_to_color(f)).save(buf, format="PNG")
            return Response(content=buf.getvalue(), media_type="image/png")
        mag = np.sqrt((f ** 2).sum(-1))
        return JSONResponse({
            "shape": list(f.shape),
            "mean_magnitude": float(mag.mean()),
            "max_magnitude": float(mag.max()),
            "mean_u": float(f[..., 0].mean()),
            "mean_v": float(f[..., 1].mean()),
        })

    return app


def serve_from_config(cfg, checkpoint: str | None, host: str, port: int):
    import uvicorn

    from .losses.unsup import DATASET_MEANS
    from .models import build_model

    model, flow_scales, _ = build_model(cfg.model, act=cfg.activation)
    device = torch.device("cuda" if cfg.device == "cuda"
                          and torch.cuda.is_available() else "cpu")
    model.to(device)
    if device.type == "cuda" and cfg.channels_last:
        model.to(memory_format=torch.channels_last)
    if checkpoint:
        state = torch.load(checkpoint, map_location=device,
                           weights_only=False)
        model.load_state_dict(state["model"])
    mean = DATASET_MEANS.get(cfg.dataset, (127.5, 127.5, 127.5))
    app = create_app(model, flow_scales, mean, device, cfg.dataset,
                     cfg.precision, cfg.eval_mult,
                     tuple(cfg.eval_clip) if cfg.eval_clip else None)
    uvicorn.run(app, host=host, port=port)
