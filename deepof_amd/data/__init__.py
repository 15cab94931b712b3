from .synthetic import SyntheticFlowDataset, SyntheticActionDataset
from .flying_chairs import FlyingChairsDataset
from .sintel import SintelDataset
from .ucf101 import UCF101Dataset
from .loader import build_dataloader

__all__ = [
    "SyntheticFlowDataset",
    "SyntheticActionDataset",
    "FlyingChairsDataset",
    "SintelDataset",
    "UCF101Dataset",
    "build_dataloader",
]
