"""deepof_amd — MI355X-native optical-flow training framework.

A from-scratch re-design of the capabilities of bryanyzhu/deepOF (the
TensorFlow "Guided Optical Flow Learning" codebase) for AMD Instinct
MI355X (gfx950): PyTorch-ROCm as the tensor/autograd substrate,
hand-written CDNA4 HIP kernels for the hot ops (bilinear flow warp,
fused Charbonnier photometric + smoothness losses, correlation cost
volume, fused conv+ELU, multi-tensor Adam), and RCCL over xGMI for
data-parallel training.

Capability parity targets (reference file:line cites live in each
module's docstring): FlowNetS/FlowNetC/VGG16/Inception-v3 encoders with
the 6-scale flow decoder, unsupervised photometric warp loss, guided
proxy-label loss, FlyingChairs/Sintel/UCF101 data plane, `.flo`
Middlebury I/O, AEE evaluation protocol and flow visualization.
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401
