from .unsup import (MultiFrameUnsupLoss, MultiScaleUnsupLoss,
                    preprocess_images)
from .guided import MultiScaleGuidedLoss
from .perceptual import PerceptualWarpLoss

__all__ = ["MultiScaleUnsupLoss", "MultiFrameUnsupLoss",
           "MultiScaleGuidedLoss", "PerceptualWarpLoss",
           "preprocess_images"]
