from .launch import init_distributed, shard_dataset
from .ddp import DataParallelGrads
