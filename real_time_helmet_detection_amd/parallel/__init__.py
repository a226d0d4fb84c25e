from .ddp import BucketedDataParallel
from .launch import (distributed_device_train, init_process_group_from_args,
                     launched_from_torchrun)

__all__ = ['BucketedDataParallel', 'distributed_device_train',
           'init_process_group_from_args', 'launched_from_torchrun']
