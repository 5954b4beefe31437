from .bucket_ddp import BucketedDDP

__all__ = ["BucketedDDP"]
