from .flownet import FlowNetS, FlowNetC
from .vgg16 import VGG16Flow
from .inception import InceptionFlow
from .ucf101 import STBaseline, STSingle
from .registry import build_model, MODEL_REGISTRY

__all__ = ["FlowNetS", "FlowNetC", "VGG16Flow", "InceptionFlow",
           "STSingle", "STBaseline", "build_model", "MODEL_REGISTRY"]
