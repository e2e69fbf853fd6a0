from .ddp import DistributedDataParallelSE3, setup_distributed
from .zero import Zero1Optimizer
