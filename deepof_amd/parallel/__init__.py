from .ddp import BucketedDataParallel, init_distributed, is_distributed

__all__ = ["BucketedDataParallel", "init_distributed", "is_distributed"]
