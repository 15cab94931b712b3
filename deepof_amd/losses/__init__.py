from .unsup import (MultiFrameUnsupLoss, MultiScaleUnsupLoss,
                    preprocess_images)
from .guided import MultiScaleGuidedLoss

__all__ = ["MultiScaleUnsupLoss", "MultiFrameUnsupLoss",
           "MultiScaleGuidedLoss", "preprocess_images"]
