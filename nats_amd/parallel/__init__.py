from .ddp import DataParallelGrads, init_distributed, distributed_info

__all__ = ["DataParallelGrads", "init_distributed", "distributed_info"]
