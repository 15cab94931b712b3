"""Single dataclass config with YAML + CLI overrides.

Replaces the reference's scattered tf.app.flags / hardcoded
hyper-params; every field of the §2.5 config matrix is a
named field here.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional

import yaml


@dataclass
class Config:
    # experiment
    run_name: str = "run"
    log_dir: str = "logs"
    seed: int = 0

    # data
    dataset: str = "synthetic"           # synthetic|flying_chairs|sintel|ucf101
    data_dir: str = ""
    image_size: tuple = (384, 512)       # (H, W)
    crop_size: Optional[tuple] = None
    batch_size: int = 8
    num_workers: int = 4
    time_step: int = 2                   # >2 = Sintel multi-frame volumes
    sintel_pass: str = "clean"

    # model
    model: str = "flownets"              # flownets|flownetc|vgg16|inception_v3
    activation: str = "elu"

    augment: bool = False                # GPU geo+photo augmentation
    vgg_init: str = ""                   # path to vgg16_weights.npz warm start

    # loss (SURVEY §2.5 defaults: chairs FlowNetS v0)
    epsilon: float = 1e-4
    alpha_c: float = 0.25
    alpha_s: float = 0.37
    lambda_smooth: float = 1.0
    edge_aware: bool = False             # loss_interp_bk image-gradient masks
    loss_weights: Optional[list] = None  # default from model registry
    guided: bool = False                 # add proxy-label supervision
    guided_weight: float = 1.0
    photo_weight: float = 1.0
    perceptual_weight: float = 0.0       # VGG feature-space warp loss

    # optimization
    lr: float = 1.6e-5
    lr_decay: float = 0.5
    epochs_per_decay: int = 18
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    weight_decay: float = 0.0
    max_epochs: int = 110
    grad_clip: float = 0.0
    grad_accumulation: int = 1           # micro-batches per optimizer step

    # runtime
    precision: str = "bf16"              # bf16|fp32
    device: str = "cuda"
    channels_last: bool = True   # NHWC: MIOpen NCHW-bf16 falls back to naive kernels
    save_interval_epochs: int = 5
    log_interval: int = 50
    eval_interval_epochs: int = 1
    resume: bool = True

    nan_restart_limit: int = 3           # auto-restarts from ckpt on NaN
    profile_steps: int = 0               # torch.profiler trace of N steps

    # eval post-processing overrides (None = dataset defaults; the
    # reference varies these per config: chairs pr1 x2 clip [-300,250],
    # chairs-VGG clip [-204.479, 201.3478], Sintel x amplifier 3 —
    # SURVEY §2.5 eval row)
    eval_mult: Optional[float] = None
    eval_clip: Optional[tuple] = None    # (min, max)

    # action head (UCF101 joint training)
    action_classes: int = 0              # >0 enables the action head
    action_weight: float = 1.0

    def __post_init__(self):
        if self.augment and self.guided:
            # the geometric augmentation transforms the frames the
            # network sees, but batch["flow"] stays in the original
            # frame geometry -> guided supervision would be wrong
            raise ValueError(
                "augment=True with guided=True is unsupported: the GT "
                "flow is not transformed through the geometric "
                "augmentation (disable one of them)")

    def to_dict(self):
        return dataclasses.asdict(self)

    @classmethod
    def from_yaml(cls, path: str) -> "Config":
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        return cls.from_dict(data)

    @classmethod
    def from_dict(cls, data: dict) -> "Config":
        names = {f.name for f in dataclasses.fields(cls)}
        unknown = set(data) - names
        if unknown:
            raise ValueError(f"unknown config keys: {sorted(unknown)}")
        cfg = cls(**data)
        if isinstance(cfg.image_size, list):
            cfg.image_size = tuple(cfg.image_size)
        if isinstance(cfg.crop_size, list):
            cfg.crop_size = tuple(cfg.crop_size)
        return cfg

    def apply_overrides(self, overrides: list[str]) -> "Config":
        """key=value overrides (YAML-parsed values)."""
        data = self.to_dict()
        for ov in overrides:
            if "=" not in ov:
                raise ValueError(f"override must be key=value, got {ov!r}")
            k, v = ov.split("=", 1)
            if k not in data:
                raise ValueError(f"unknown config key {k!r}")
            parsed = yaml.safe_load(v)
            # YAML 1.1 reads dotless scientific notation ("1e-07") as a
            # STRING; coerce to the field's numeric type so e.g.
            # lr=1e-07 does not silently poison the optimizer
            cur = data[k]
            if isinstance(parsed, str) and isinstance(cur, (int, float)) \
                    and not isinstance(cur, bool):
                try:
                    parsed = type(cur)(float(parsed)) \
                        if isinstance(cur, int) else float(parsed)
                except ValueError:
                    pass
            data[k] = parsed
        return Config.from_dict(data)
