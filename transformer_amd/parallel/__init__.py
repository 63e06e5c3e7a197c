from .ddp import BucketedDataParallel, init_distributed  # noqa: F401
