from .ddp import BucketedDataParallel  # noqa: F401
