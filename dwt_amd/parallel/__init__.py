from .ddp import BucketedDataParallel, init_distributed_from_env, is_distributed

__all__ = ["BucketedDataParallel", "init_distributed_from_env", "is_distributed"]
