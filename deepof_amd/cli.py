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
    p_eval.add_argument("overrides", nargs="*")

    args = parser.parse_args(argv)
    cfg = Config.from_yaml(args.config) if args.config else Config()
    cfg = cfg.apply_overrides(args.overrides)

    if args.command == "train":
        from .engine import Trainer

        Trainer(cfg).fit(max_steps=args.max_steps)
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
                           cfg.dataset)
        print(f"AEE: {aee:.4f}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
