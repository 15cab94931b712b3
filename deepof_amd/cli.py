"""CLI: `python -m deepof_amd train|eval ...`

Mirrors the reference entrypoints (version1/deepOF.py:12-37 argparse ->
opts -> train(opts)) with a dataclass/YAML config and key=value
overrides.
"""

from __future__ import annotations

import argparse
import sys

from .config import Config


def main(argv=None):
    parser = argparse.ArgumentParser(prog="deepof_amd")
    sub = parser.add_subparsers(dest="command", required=True)

    p_train = sub.add_parser("train", help="train a flow model")
    p_train.add_argument("--config", type=str, default=None,
                         help="YAML config path")
    p_train.add_argument("--max-steps", type=int, default=None)
    p_train.add_argument("overrides", nargs="*",
                         help="key=value config overrides")

    p_eval = sub.add_parser("eval", help="evaluate AEE from a checkpoint")
    p_eval.add_argument("--config", type=str, default=None)
    p_eval.add_argument("--checkpoint", type=str, required=False)
    p_eval.add_argument("--dump-dir", type=str, default=None,
                        help="write flow color maps / .flo / warped frames")
    p_eval.add_argument("overrides", nargs="*")

    p_serve = sub.add_parser("serve",
                             help="HTTP flow inference service (FastAPI)")
    p_serve.add_argument("--config", type=str, default=None)
    p_serve.add_argument("--checkpoint", type=str, default=None)
    p_serve.add_argument("--host", type=str, default="127.0.0.1")
    p_serve.add_argument("--port", type=int, default=8000)
    p_serve.add_argument("overrides", nargs="*")

    p_infer = sub.add_parser("infer", help="predict flow for an image pair")
    p_infer.add_argument("--config", type=str, default=None)
    p_infer.add_argument("--checkpoint", type=str, required=True)
    p_infer.add_argument("--img1", type=str, required=True)
    p_infer.add_argument("--img2", type=str, required=True)
    p_infer.add_argument("--out", type=str, default="flow_out",
                         help="output prefix (.flo and .jpg written)")
    p_infer.add_argument("overrides", nargs="*")

    args = parser.parse_args(argv)
    cfg = Config.from_yaml(args.config) if args.config else Config()
    cfg = cfg.apply_overrides(args.overrides)

    if args.command == "train":
        from .engine import Trainer

        Trainer(cfg).fit(max_steps=args.max_steps)
    elif args.command == "serve":
        from .serve import serve_from_config

        serve_from_config(cfg, args.checkpoint, args.host, args.port)
    elif args.command == "eval":
        import torch

        from .data import build_dataloader
        from .engine.evaluator import evaluate_aee
        from .engine.trainer import build_datasets
        from .losses.unsup import DATASET_MEANS
        from .models import build_model

        model, flow_scales, _ = build_model(cfg.model, act=cfg.activation)
        device = torch.device("cuda" if cfg.device == "cuda"
                              and torch.cuda.is_available() else "cpu")
        model.to(device)
        if args.checkpoint:
            state = torch.load(args.checkpoint, map_location=device,
                               weights_only=False)
            model.load_state_dict(state["model"])
        _, val_ds = build_datasets(cfg)
        loader = build_dataloader(val_ds, cfg.batch_size, shuffle=False,
                                  drop_last=False, num_workers=2)
        mean = DATASET_MEANS.get(cfg.dataset, (127.5, 127.5, 127.5))
        aee = evaluate_aee(model, loader, mean, flow_scales[0], device,
                           cfg.dataset, dump_dir=args.dump_dir,
                           mult=cfg.eval_mult,
                           clip=tuple(cfg.eval_clip) if cfg.eval_clip
                           else None)
        print(f"AEE: {aee:.4f}")
    elif args.command == "infer":
        import numpy as np
        import torch
        from PIL import Image

        from .data.image import load_image, to_chw
        from .engine.evaluator import predict_flow
        from .losses.unsup import DATASET_MEANS
        from .models import build_model
        from .utils import flow_to_color, write_flo

        model, flow_scales, _ = build_model(cfg.model, act=cfg.activation)
        device = torch.device("cuda" if cfg.device == "cuda"
                              and torch.cuda.is_available() else "cpu")
        model.to(device).eval()
        state = torch.load(args.checkpoint, map_location=device,
                           weights_only=False)
        model.load_state_dict(state["model"] if "model" in state else state)

        img1 = torch.from_numpy(to_chw(load_image(args.img1))).unsqueeze(0)
        img2 = torch.from_numpy(to_chw(load_image(args.img2))).unsqueeze(0)
        mean = DATASET_MEANS.get(cfg.dataset, (127.5, 127.5, 127.5))
        with torch.no_grad():
            pred = predict_flow(model, img1.to(device), img2.to(device),
                                mean, flow_scales[0], cfg.dataset,
                                gt_size=tuple(img1.shape[-2:]))
        flow = pred[0].permute(1, 2, 0).cpu().numpy()
        write_flo(args.out + ".flo", flow)
        Image.fromarray(flow_to_color(flow)).save(args.out + ".jpg")
        print(f"wrote {args.out}.flo and {args.out}.jpg "
              f"(|f| max {np.abs(flow).max():.2f})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
