from .ddp import FlatDDP
from .sync_bn import SyncBatchNorm, convert_sync_batchnorm
from .launch import init_multiprocessing_and_cuda, launch, cleanup

__all__ = ["FlatDDP", "SyncBatchNorm", "convert_sync_batchnorm",
           "init_multiprocessing_and_cuda", "launch", "cleanup"]
