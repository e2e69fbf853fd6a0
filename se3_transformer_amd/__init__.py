"""se3_transformer_amd — MI355X-native SE(3)-equivariant transformer framework.

A from-scratch implementation with the public API and capabilities of
lucidrains/se3-transformer-pytorch (v0.9.0), built for AMD Instinct MI355X
(gfx950): PyTorch-ROCm eager oracle + hand-written CDNA4 HIP kernels for the
hot ops + RCCL-over-xGMI data parallelism.
"""
from .models import SE3Transformer, Fiber, FiberEl
from .models import (AttentionBlockSE3, AttentionSE3, ConvSE3, EGNN,
                     EGnnNetwork, FeedForwardBlockSE3, FeedForwardSE3,
                     LinearSE3, NormSE3, OneHeadedKVAttentionSE3, PairwiseConv,
                     RadialFunc, ResidualSE3)
from .ops import get_basis

__version__ = '0.1.0'
