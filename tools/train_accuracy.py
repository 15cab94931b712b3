"""Produce the first measured accuracy number: train
FlowNetS unsupervised on the synthetic dataset (whose GT flow is an
exact photometric minimum, data/synthetic.py) and report AEE through
the full reference eval protocol (pr1 x2, clip, resize —
/root/reference/flyingChairsTrain.py:264-296).

    gpurun -- 'python tools/train_accuracy.py --epochs 40'

Writes gpurun_out/accuracy_run/{metrics.jsonl,result.json}.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, ".")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=40)
    ap.add_argument("--minutes", type=float, default=6.0)
    ap.add_argument("--height", type=int, default=96)
    ap.add_argument("--width", type=int, default=128)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--lr", type=float, default=1e-4)
    ap.add_argument("--guided", action="store_true",
                    help="add proxy-label supervision (BASELINE configs[3] "
                         "guided + warp-photometric loss)")
    args = ap.parse_args()

    import torch

    from deepof_amd.config import Config
    from deepof_amd.data import SyntheticFlowDataset, build_dataloader
    from deepof_amd.engine import Trainer
    from deepof_amd.engine.evaluator import evaluate_aee

    os.makedirs("gpurun_out", exist_ok=True)
    cfg = Config.from_dict(dict(
        dataset="synthetic", image_size=(args.height, args.width),
        batch_size=args.batch, num_workers=8, model="flownets",
        precision="bf16", device="cuda", log_dir="gpurun_out",
        run_name="accuracy_run", lr=args.lr, epochs_per_decay=15,
        max_epochs=args.epochs, log_interval=50, eval_interval_epochs=5,
        save_interval_epochs=10, resume=False, seed=0,
        guided=args.guided, guided_weight=1.0,
    ))
    tr = Trainer(cfg)
    t0 = time.time()
    deadline = t0 + args.minutes * 60

    val_ds = SyntheticFlowDataset(64, args.height, args.width, seed=1)
    val_loader = build_dataloader(val_ds, args.batch, shuffle=False,
                                  num_workers=2, drop_last=False)

    aee0 = evaluate_aee(tr.raw_model, val_loader, tr.mean_bgr,
                        tr.flow_scales[0], tr.device, "synthetic")
    zero_aee = 0.0
    n = 0
    for b in val_loader:
        f = b["flow"].float()
        zero_aee += float(torch.sqrt((f ** 2).sum(1)).mean()) * f.shape[0]
        n += f.shape[0]
    zero_aee /= n
    print(f"[accuracy] AEE at init {aee0:.4f}; zero-flow baseline "
          f"{zero_aee:.4f}")

    while tr.epoch < args.epochs and time.time() < deadline:
        tr.fit(max_epochs=min(tr.epoch + 5, args.epochs))
        tr.raw_model.train()

    aee = evaluate_aee(tr.raw_model, val_loader, tr.mean_bgr,
                       tr.flow_scales[0], tr.device, "synthetic")
    result = {
        "metric": "avg EPE (synthetic val, reference protocol)",
        "aee": aee,
        "aee_init": aee0,
        "zero_flow_baseline": zero_aee,
        "epochs": tr.epoch,
        "steps": tr.global_step,
        "minutes": round((time.time() - t0) / 60, 2),
        "config": {"model": "flownets", "image": [args.height, args.width],
                   "batch": args.batch, "lr": args.lr,
                   "loss": ("guided+photometric" if args.guided else "unsupervised photometric+smoothness"),
                   "data": "synthetic (GT flow = photometric minimum)"},
    }
    print(json.dumps(result))
    with open("gpurun_out/accuracy_run/result.json", "w") as f:
        json.dump(result, f, indent=1)
    # checkpoints are ~0.5 GB (model + Adam moments): drop them so the
    # gpurun copy-back stays under its 64 MiB limit
    import glob

    for p in glob.glob("gpurun_out/accuracy_run/ckpt_*.pt"):
        os.remove(p)


if __name__ == "__main__":
    main()
