from .ddp import BucketedDDP, init_distributed

__all__ = ["BucketedDDP", "init_distributed"]
