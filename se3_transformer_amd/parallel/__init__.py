from .ddp import DistributedDataParallelSE3, setup_distributed
