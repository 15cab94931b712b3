"""Model registry: name -> (constructor, per-scale flow scales, loss weights).

Loss-weight schedules (fine -> coarse) follow the reference configs
(/root/reference SURVEY §2.5): v0 chairs FlowNetS [16,8,4,2,1,1];
version1 Flownet/Inception [9,7,5,3,3,1]; VGG [7,5,3,3,1].
"""

from __future__ import annotations

from .flownet import FLOW_SCALES, FlowNetC, FlowNetS
from .inception import INCEPTION_FLOW_SCALES, InceptionFlow
from .ucf101 import STBaseline, STSingle
from .vgg16 import VGG_FLOW_SCALES, VGG16Flow

MODEL_REGISTRY = {
    "flownets": {
        "ctor": FlowNetS,
        "flow_scales": FLOW_SCALES,
        "loss_weights": [16.0, 8.0, 4.0, 2.0, 1.0, 1.0],
    },
    "flownetc": {
        "ctor": FlowNetC,
        "flow_scales": FLOW_SCALES,
        "loss_weights": [16.0, 8.0, 4.0, 2.0, 1.0, 1.0],
    },
    "vgg16": {
        "ctor": VGG16Flow,
        "flow_scales": VGG_FLOW_SCALES,
        "loss_weights": [7.0, 5.0, 3.0, 3.0, 1.0],
    },
    "st_single": {
        "ctor": STSingle,
        "flow_scales": VGG_FLOW_SCALES,
        "loss_weights": [16.0, 8.0, 4.0, 2.0, 1.0],
    },
    "st_baseline": {
        "ctor": STBaseline,
        "flow_scales": FLOW_SCALES,
        "loss_weights": [16.0, 8.0, 4.0, 2.0, 1.0, 1.0],
    },
    "inception_v3": {
        "ctor": InceptionFlow,
        "flow_scales": INCEPTION_FLOW_SCALES,
        "loss_weights": [9.0, 7.0, 5.0, 3.0, 3.0, 1.0],
    },
}


def build_model(name: str, **kwargs):
    if name not in MODEL_REGISTRY:
        raise ValueError(f"unknown model {name!r}; have {sorted(MODEL_REGISTRY)}")
    entry = MODEL_REGISTRY[name]
    model = entry["ctor"](**kwargs)
    return model, list(entry["flow_scales"]), list(entry["loss_weights"])
