from .fiber import Fiber, FiberEl
from .core import (ConvSE3, FeedForwardBlockSE3, FeedForwardSE3, HtypesNorm,
                   LinearSE3, NormSE3, PairwiseConv, RadialFunc, ResidualSE3)
from .attention import AttentionBlockSE3, AttentionSE3, OneHeadedKVAttentionSE3
from .egnn import EGNN, EGnnNetwork
from .reversible import ReversibleSequence, SequentialSequence
from .rotary import SinusoidalEmbeddings, apply_rotary_pos_emb
from .transformer import SE3Transformer
