from .unsup import MultiScaleUnsupLoss, preprocess_images
from .guided import MultiScaleGuidedLoss

__all__ = ["MultiScaleUnsupLoss", "MultiScaleGuidedLoss", "preprocess_images"]
