"""Every shipped config must load through the validated Config path
(unknown keys / conflicting flags reject at load time)."""

import glob
import os

from deepof_amd.config import Config

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_all_shipped_configs_load():
    paths = sorted(glob.glob(os.path.join(REPO, "configs", "*.yaml")))
    assert paths, "no shipped configs found"
    for p in paths:
        cfg = Config.from_yaml(p)
        assert cfg.batch_size > 0
        assert cfg.model in ("flownets", "flownetc", "vgg16",
                             "inception_v3", "st_single", "st_baseline"), p


def test_config_model_names_buildable():
    from deepof_amd.models import build_model

    for p in sorted(glob.glob(os.path.join(REPO, "configs", "*.yaml"))):
        cfg = Config.from_yaml(p)
        kwargs = {"act": cfg.activation}
        if cfg.model == "inception_v3" and cfg.time_step > 2:
            kwargs["time_step"] = cfg.time_step
        if cfg.model in ("st_single", "st_baseline"):
            kwargs["input_hw"] = tuple(cfg.crop_size or cfg.image_size)
            kwargs["num_classes"] = cfg.action_classes or 101
        model, scales, weights = build_model(cfg.model, **kwargs)
        assert len(scales) == len(weights)
