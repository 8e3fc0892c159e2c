from .ddp import BucketedDataParallel, setup_distributed

__all__ = ["BucketedDataParallel", "setup_distributed"]
