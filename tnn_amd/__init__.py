"""tnn_amd — an MI355X-native deep-learning training framework.

A from-scratch re-design of the capabilities of tungphambasement/TNN
(reference mounted at /root/reference) for AMD Instinct MI355X (gfx950):

- PyTorch-ROCm supplies tensors/autograd; every hot op (Conv2d NHWC
  implicit-GEMM, fused BatchNorm+ReLU, Dense, pooling, losses, optimizers,
  attention) is a hand-written CDNA4 HIP kernel on MFMA with LDS-staged
  tiles (``tnn_amd/csrc/``), loaded from the in-tree extension ``tnn_amd._hip``.
- Distribution is one process per GPU over torch.distributed
  (NCCL backend == RCCL on ROCm) with pipeline parallelism over xGMI
  (``tnn_amd.parallel``), replacing the reference's TCP/RoCE
  coordinator/worker system (reference include/distributed/).
- Models are define-by-config serializable Sequentials built with a
  fluent ``LayerBuilder`` (reference include/nn/layer_builder.hpp:44).

Layout convention: image tensors are NHWC (contiguous [N, H, W, C]),
matching the reference's new layer family (reference
include/data_loading/cifar10_data_loader.hpp:37) and the natural layout
for NHWC implicit-GEMM convolution on MFMA.
"""

from .version import __version__

from . import ops
from . import nn
from . import data
from . import utils
from . import models
from . import parallel

__all__ = ["ops", "nn", "data", "utils", "models", "parallel", "__version__"]
