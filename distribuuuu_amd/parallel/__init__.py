"""Parallelism layer: xGMI-tuned DDP and RCCL SyncBatchNorm."""

from .ddp import DistributedDataParallel  # noqa: F401
from .syncbn import SyncBatchNorm, convert_sync_batchnorm  # noqa: F401
