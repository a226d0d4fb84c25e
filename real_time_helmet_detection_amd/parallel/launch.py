"""Process-per-GPU launch for single-node multi-GPU training.

Mirrors the reference's mp.spawn semantics (/root/reference/train.py:23-30):
``world_size = ngpus_per_node * args.world_size``, rank =
``args.rank * ngpus_per_node + local_rank``, rendezvous at ``--dist-url``.

Additionally supports torchrun-style env launch (RANK/LOCAL_RANK/WORLD_SIZE
set) — the bench driver uses ``python -m torch.distributed.run`` — in which
case we do NOT spawn and just run the worker in-process.

Backend: 'nccl' (RCCL over xGMI) on GPU, 'gloo' on CPU (multi-process CPU
tests run the identical worker path).
"""

import os
import datetime

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def launched_from_torchrun():
    return 'RANK' in os.environ and 'WORLD_SIZE' in os.environ


def init_process_group_from_args(args, rank, world_size):
    backend = args.dist_backend
    if not torch.cuda.is_available() and backend == 'nccl':
        backend = 'gloo'
    # under torchrun, rendezvous on its MASTER_ADDR/PORT (env://) instead
    # of the --dist-url default, which may name a different port
    init_method = 'env://' if launched_from_torchrun() else args.dist_url
    dist.init_process_group(
        backend=backend,
        init_method=init_method,
        world_size=world_size,
        rank=rank,
        timeout=datetime.timedelta(seconds=300),
    )
    return backend


def distributed_device_train(args):
    """Entry used by main.py for training (reference train.py:23-30)."""
    from ..engine.trainer import distributed_worker

    if launched_from_torchrun():
        # torchrun already created one process per GPU
        local_rank = int(os.environ.get('LOCAL_RANK', '0'))
        ngpus = int(os.environ.get('LOCAL_WORLD_SIZE',
                                   os.environ['WORLD_SIZE']))
        distributed_worker(local_rank, ngpus, args, env_launch=True)
        return

    ngpus_per_node = torch.cuda.device_count() if torch.cuda.is_available() \
        else 1
    args.world_size = ngpus_per_node * args.world_size
    if args.world_size == 1:
        distributed_worker(0, 1, args)
    else:
        mp.spawn(distributed_worker, nprocs=ngpus_per_node,
                 args=(ngpus_per_node, args))
