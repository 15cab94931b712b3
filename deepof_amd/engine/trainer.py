"""Training runtime: dataset/model dispatch, epoch loop, LR schedule,
mixed precision, DDP, NaN guard, checkpointing, JSONL metrics.

Parity targets: version1/trainOF.py:16-219 (dispatch, Adam, epoch loop,
0.5x LR decay every N epochs, checkpoint save/restore) with the
reference's training-time invariants kept (NaN divergence guard,
flyingChairsTrain.py:203).  Checkpoints carry model + Adam moments +
epoch, like tf.train.Saver captured all variables.
"""

from __future__ import annotations

import json
import math
import os
import time

import torch

from ..config import Config
from ..data import (FlyingChairsDataset, SintelDataset, SyntheticFlowDataset,
                    UCF101Dataset, build_dataloader)
from ..losses import MultiScaleGuidedLoss, MultiScaleUnsupLoss
from ..losses.unsup import DATASET_MEANS
from ..models import build_model
from ..parallel import BucketedDataParallel, init_distributed
from .evaluator import evaluate_aee
from .optim import FusedAdam


def build_datasets(cfg: Config):
    h, w = cfg.crop_size or cfg.image_size
    if cfg.dataset == "synthetic":
        train = SyntheticFlowDataset(4096, h, w, seed=cfg.seed)
        val = SyntheticFlowDataset(64, h, w, seed=cfg.seed + 1)
    elif cfg.dataset == "flying_chairs":
        train = FlyingChairsDataset(cfg.data_dir, "train", image_size=(h, w))
        val = FlyingChairsDataset(cfg.data_dir, "val", image_size=(h, w))
    elif cfg.dataset == "sintel":
        train = SintelDataset(cfg.data_dir, "train", cfg.sintel_pass,
                              cfg.time_step, image_size=(h, w))
        val = SintelDataset(cfg.data_dir, "val", cfg.sintel_pass,
                            cfg.time_step, image_size=(h, w))
    elif cfg.dataset == "ucf101":
        train = UCF101Dataset(cfg.data_dir, "train", image_size=(h, w))
        val = UCF101Dataset(cfg.data_dir, "test", image_size=(h, w))
    else:
        raise ValueError(f"unknown dataset {cfg.dataset!r}")
    return train, val


class Trainer:
    def __init__(self, cfg: Config):
        self.cfg = cfg
        self.rank, self.local_rank, self.world = init_distributed()
        self.device = torch.device(
            f"cuda:{self.local_rank}" if cfg.device == "cuda"
            and torch.cuda.is_available() else "cpu"
        )
        torch.manual_seed(cfg.seed + self.rank)

        model_kwargs = {"act": cfg.activation}
        if cfg.model == "inception_v3" and cfg.time_step > 2:
            model_kwargs["time_step"] = cfg.time_step
        if cfg.model in ("st_single", "st_baseline"):
            model_kwargs["input_hw"] = tuple(cfg.crop_size or cfg.image_size)
            model_kwargs["num_classes"] = cfg.action_classes or 101
        self.model, self.flow_scales, default_w = build_model(
            cfg.model, **model_kwargs)
        if cfg.vgg_init:
            from ..utils.warmstart import load_vgg16_npz

            enc = getattr(self.model, "encoder", None) or getattr(
                self.model, "spatial", None)
            n = load_vgg16_npz(enc, cfg.vgg_init)
            if self.rank == 0:
                print(f"[deepof] VGG16 warm start: {n} tensors from "
                      f"{cfg.vgg_init}")
        self.loss_weights = cfg.loss_weights or default_w
        self.model.to(self.device)
        if cfg.channels_last:
            self.model.to(memory_format=torch.channels_last)

        mean = DATASET_MEANS.get(cfg.dataset, (127.5, 127.5, 127.5))
        self.mean_bgr = mean
        self.unsup_loss = MultiScaleUnsupLoss(
            self.flow_scales, self.loss_weights, mean,
            cfg.epsilon, cfg.alpha_c, cfg.alpha_s, cfg.lambda_smooth,
            edge_aware=cfg.edge_aware,
        )
        self.guided_loss = (
            MultiScaleGuidedLoss(self.flow_scales, self.loss_weights)
            if cfg.guided else None
        )
        self.perceptual_loss = None
        if cfg.perceptual_weight > 0:
            from ..losses import PerceptualWarpLoss

            self.perceptual_loss = PerceptualWarpLoss().to(self.device)

        if self.world > 1:
            self.model = BucketedDataParallel(self.model)
        self.optimizer = FusedAdam(
            self.model.parameters(), lr=cfg.lr,
            betas=(cfg.adam_beta1, cfg.adam_beta2), eps=cfg.adam_eps,
            weight_decay=cfg.weight_decay,
        )
        self.epoch = 0
        self.global_step = 0
        self._micro_step = 0
        self.best_aee = float("inf")

        self.run_dir = os.path.join(cfg.log_dir, cfg.run_name)
        if self.rank == 0:
            os.makedirs(self.run_dir, exist_ok=True)
        self._metrics_f = None

        if cfg.resume:
            self.try_resume()

    # -- plumbing ---------------------------------------------------------
    @property
    def raw_model(self):
        return self.model.module if isinstance(
            self.model, BucketedDataParallel) else self.model

    def log_metrics(self, record: dict):
        if self.rank != 0:
            return
        if self._metrics_f is None:
            self._metrics_f = open(
                os.path.join(self.run_dir, "metrics.jsonl"), "a")
        self._metrics_f.write(json.dumps(record) + "\n")
        self._metrics_f.flush()

    def checkpoint_path(self, epoch=None):
        name = f"ckpt_{epoch:04d}.pt" if epoch is not None else "ckpt_last.pt"
        return os.path.join(self.run_dir, name)

    def save_checkpoint(self):
        if self.rank != 0:
            return
        state = {
            "model": self.raw_model.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "epoch": self.epoch,
            "global_step": self.global_step,
            "config": self.cfg.to_dict(),
        }
        tmp = self.checkpoint_path() + ".tmp"
        torch.save(state, tmp)
        os.replace(tmp, self.checkpoint_path())

    def try_resume(self):
        path = self.checkpoint_path()
        if not os.path.exists(path):
            return False
        state = torch.load(path, map_location=self.device,
                           weights_only=False)
        self.raw_model.load_state_dict(state["model"])
        self.optimizer.load_state_dict(state["optimizer"])
        self.epoch = state["epoch"]
        self.global_step = state["global_step"]
        if self.rank == 0:
            print(f"[deepof] resumed from {path} at epoch {self.epoch}")
        return True

    def current_lr(self):
        decays = self.epoch // self.cfg.epochs_per_decay
        return self.cfg.lr * (self.cfg.lr_decay**decays)

    def _zero_grads(self):
        if isinstance(self.model, BucketedDataParallel):
            self.model.zero_grad_buckets()
        else:
            # keep grads allocated on GPU so the fused-Adam chunk table's
            # cached device pointers stay valid (set_to_none=True would
            # reallocate every grad each backward -> table rebuild/step)
            self.optimizer.zero_grad(
                set_to_none=self.device.type != "cuda")

    # -- the step ---------------------------------------------------------
    def train_step(self, batch, log_scales: bool = False) -> dict:
        """One micro-batch.  log_scales=True additionally reports the
        per-scale loss components (the reference's display-interval
        print, flyingChairsTrain.py:183-201) — off by default since
        each component read is a device sync."""
        cfg = self.cfg
        from ..losses.unsup import preprocess_images

        use_bf16 = cfg.precision == "bf16" and self.device.type == "cuda"
        parts = {}

        if "volume" in batch:  # Sintel multi-frame mode
            vol = batch["volume"].to(self.device, non_blocking=True)
            from ..losses import MultiFrameUnsupLoss

            if not hasattr(self, "_mf_loss"):
                self._mf_loss = MultiFrameUnsupLoss(
                    self.flow_scales, self.loss_weights, self.mean_bgr,
                    cfg.epsilon, cfg.alpha_c, cfg.alpha_s, cfg.lambda_smooth)
            mean = torch.as_tensor(self.mean_bgr, dtype=torch.float32,
                                   device=vol.device)
            T = vol.shape[1] // 3
            x = (vol.float() - mean.repeat(T).view(1, -1, 1, 1)) / 255.0
            with torch.autocast("cuda", dtype=torch.bfloat16,
                                enabled=use_bf16):
                flows = self.model(x)
            res = self._mf_loss(flows, vol)
            total = res["total"]
            parts["unsup"] = float(total.detach())
            if log_scales:
                parts["scale_losses"] = [
                    round(float(s.detach()), 6) for s in res["scales"]]
        else:
            img1 = batch["img1"].to(self.device, non_blocking=True)
            img2 = batch["img2"].to(self.device, non_blocking=True)
            geo1, geo2 = img1.float(), img2.float()
            net1, net2 = geo1, geo2
            if cfg.augment:
                from ..utils.augment import augment_pair

                geo1, geo2, net1, net2 = augment_pair(geo1, geo2)
            x1 = preprocess_images(net1, self.mean_bgr)
            x2 = preprocess_images(net2, self.mean_bgr)
            x = torch.cat([x1, x2], dim=1)
            if cfg.channels_last:
                x = x.to(memory_format=torch.channels_last)

            with torch.autocast("cuda", dtype=torch.bfloat16,
                                enabled=use_bf16):
                out = self.model(x)
            logits = None
            flows = out
            if isinstance(out, tuple):  # joint flow+action models
                flows, logits = out
            res = self.unsup_loss(flows, geo1, geo2)
            total = cfg.photo_weight * res["total"]
            parts["unsup"] = float(res["total"].detach())
            if log_scales:
                parts["scale_losses"] = [
                    round(float(s["total"].detach()), 6)
                    for s in res["scales"]]
            if self.guided_loss is not None and "flow" in batch:
                gt = batch["flow"].to(self.device, non_blocking=True)
                g = self.guided_loss(flows, gt)
                total = total + cfg.guided_weight * g["total"]
                parts["guided"] = float(g["total"].detach())
            if self.perceptual_loss is not None:
                import torch.nn.functional as TF

                H, W = geo1.shape[-2:]
                flow_full = TF.interpolate(
                    flows[0].float(), size=(H, W), mode="bilinear",
                    align_corners=False) * (self.flow_scales[0] * 2.0)
                from ..losses.unsup import preprocess_images as _pp

                pl = self.perceptual_loss(flow_full,
                                          _pp(geo1, self.mean_bgr),
                                          _pp(geo2, self.mean_bgr))
                total = total + cfg.perceptual_weight * pl
                parts["perceptual"] = float(pl.detach())
            if logits is not None and "label" in batch:
                labels = batch["label"].to(self.device, non_blocking=True)
                ce = torch.nn.functional.cross_entropy(logits.float(), labels)
                total = total + cfg.action_weight * ce
                parts["action_ce"] = float(ce.detach())
                parts["action_acc"] = float(
                    (logits.argmax(1) == labels).float().mean())

        # gradient accumulation: micro-batches sum into .grad (and the
        # DDP flat buckets); the all-reduce and optimizer step run only
        # on the boundary micro-batch
        accum = max(1, cfg.grad_accumulation)
        boundary = (self._micro_step + 1) % accum == 0
        if accum > 1:
            total = total / accum
        if isinstance(self.model, BucketedDataParallel):
            self.model.accumulate_only = not boundary
        total.backward()
        self._micro_step += 1

        loss_val = float(total.detach()) * accum
        if math.isnan(loss_val) or math.isinf(loss_val):
            raise FloatingPointError(
                f"Model diverged (loss={loss_val}) at step {self.global_step}"
            )

        if boundary:
            if isinstance(self.model, BucketedDataParallel):
                self.model.finish_gradient_sync()
            if cfg.grad_clip > 0:
                torch.nn.utils.clip_grad_norm_(self.model.parameters(),
                                               cfg.grad_clip)
            for g in self.optimizer.param_groups:
                g["lr"] = self.current_lr()
            self.optimizer.step()
            self._zero_grads()
        self.global_step += 1
        parts["total"] = loss_val
        return parts

    # -- the loop ---------------------------------------------------------
    def fit(self, max_epochs=None, max_steps=None):
        cfg = self.cfg
        train_ds, val_ds = build_datasets(cfg)
        loader = build_dataloader(
            train_ds, cfg.batch_size, num_workers=cfg.num_workers,
            distributed=self.world > 1, seed=cfg.seed,
        )
        val_loader = build_dataloader(
            val_ds, cfg.batch_size, shuffle=False,
            num_workers=min(cfg.num_workers, 2), drop_last=False,
        )
        max_epochs = max_epochs or cfg.max_epochs
        steps_done = 0
        nan_restarts = 0
        profiler = None
        if cfg.profile_steps > 0 and self.rank == 0:
            profiler = torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CPU,
                            torch.profiler.ProfilerActivity.CUDA],
                schedule=torch.profiler.schedule(
                    wait=1, warmup=2, active=cfg.profile_steps, repeat=1),
                on_trace_ready=torch.profiler.tensorboard_trace_handler(
                    self.run_dir),
            )
            profiler.start()
        while self.epoch < max_epochs:
            if hasattr(loader, "sampler") and hasattr(loader.sampler, "set_epoch"):
                loader.sampler.set_epoch(self.epoch)
            t0 = time.time()
            n_imgs = 0
            batches = loader
            if self.device.type == "cuda":
                from ..data.loader import CudaPrefetcher

                batches = CudaPrefetcher(loader, self.device)
            for i, batch in enumerate(batches):
                try:
                    parts = self.train_step(
                        batch,
                        log_scales=(self.global_step + 1)
                        % cfg.log_interval == 0)
                except FloatingPointError as e:
                    # divergence guard (reference asserts and dies,
                    # flyingChairsTrain.py:203); here: restart from the
                    # latest checkpoint, bounded retries.  Under DDP a
                    # per-rank restart would desynchronize the bucket
                    # all-reduces (other ranks keep training), so
                    # distributed runs fail fast instead.
                    if self.world > 1:
                        raise
                    nan_restarts += 1
                    if (nan_restarts > cfg.nan_restart_limit
                            or not self.try_resume()):
                        raise
                    self._zero_grads()
                    self._micro_step = 0  # restart accumulation cleanly
                    if isinstance(self.model, BucketedDataParallel):
                        self.model.accumulate_only = False
                    print(f"[deepof] {e}; restarted from checkpoint "
                          f"({nan_restarts}/{cfg.nan_restart_limit})")
                    break
                # batch size from whichever key this mode carries
                # ("img1" for pairs, "volume" for Sintel multi-frame)
                first = batch.get("img1", batch.get("volume"))
                n_imgs += first.shape[0] * self.world
                steps_done += 1
                if self.global_step % cfg.log_interval == 0:
                    dt = time.time() - t0
                    rec = {
                        "step": self.global_step, "epoch": self.epoch,
                        "lr": self.current_lr(),
                        "imgs_per_sec": n_imgs / max(dt, 1e-9), **parts,
                    }
                    self.log_metrics(rec)
                    if self.rank == 0:
                        print(f"[deepof] {rec}")
                if profiler is not None:
                    profiler.step()
                    if steps_done >= cfg.profile_steps + 4:
                        profiler.stop()
                        profiler = None
                if max_steps is not None and steps_done >= max_steps:
                    self.save_checkpoint()
                    return
            self.epoch += 1
            if (self.rank == 0 and val_ds is not None
                    and self.epoch % cfg.eval_interval_epochs == 0):
                sample = val_ds[0]
                if "flow" in sample:
                    aee = evaluate_aee(
                        self.raw_model, val_loader, self.mean_bgr,
                        self.flow_scales[0], self.device, cfg.dataset,
                        mult=cfg.eval_mult,
                        clip=tuple(cfg.eval_clip) if cfg.eval_clip else None,
                    )
                    self.log_metrics({"epoch": self.epoch, "aee": aee})
                    print(f"[deepof] ***Test: epoch {self.epoch} "
                          f"AEE {aee:.4f}")
                    if aee < self.best_aee:
                        self.best_aee = aee
                        state = {
                            "model": self.raw_model.state_dict(),
                            "epoch": self.epoch, "aee": aee,
                            "config": self.cfg.to_dict(),
                        }
                        tmp = os.path.join(self.run_dir, "ckpt_best.pt.tmp")
                        torch.save(state, tmp)
                        os.replace(tmp, os.path.join(self.run_dir,
                                                     "ckpt_best.pt"))
                if "label" in sample and cfg.action_classes > 0:
                    from .evaluator import evaluate_accuracy

                    acc = evaluate_accuracy(self.raw_model, val_loader,
                                            self.mean_bgr, self.device)
                    self.log_metrics({"epoch": self.epoch, "accuracy": acc})
                    print(f"[deepof] ***Test: epoch {self.epoch} "
                          f"accuracy {acc:.4f}")
            if self.epoch % cfg.save_interval_epochs == 0:
                self.save_checkpoint()
        self.save_checkpoint()
