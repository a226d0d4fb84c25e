"""RCCL all-reduce microbenchmark vs the xGMI per-link bound.

SURVEY.md §7 hard-part 4: the MI355X node is fully connected — each GPU has
7 point-to-point xGMI links of ~153 GB/s. A ring all-reduce moves
2(n-1)/n * S bytes over ONE link per GPU, so the ring bus bandwidth is
bounded by ~153 GB/s regardless of n; RCCL may also pick direct/tree
algorithms that aggregate links. This sweep measures algbw/busbw across
message sizes and dtypes so the DDP bucket size (--bucket-cap-mb) can be
set from DATA, not guesswork.

Launch (any world size the node offers; one rank per GPU):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port 29600 tools/allreduce_bench.py

Rank 0 prints one JSON line per (dtype, size) with us/call, algbw and
busbw GB/s; copy the output into profiles/allreduce_sweep_rNN.json.
No GPU -> falls back to gloo on CPU (plumbing test only).
"""

import json
import os
import sys
import time

import torch
import torch.distributed as dist

SIZES_MB = [0.25, 0.5, 1, 2, 4, 5, 8, 16, 32, 64]
DTYPES = {'fp32': torch.float32, 'bf16': torch.bfloat16}
LINK_GBPS = 153.0  # xGMI p2p per-link, per direction


def bench_one(tensor, iters, warmup, device):
    for _ in range(warmup):
        dist.all_reduce(tensor)
    if device.type == 'cuda':
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        dist.all_reduce(tensor)
    if device.type == 'cuda':
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    dist.barrier()
    return elapsed / iters


def main():
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local_rank = int(os.environ.get('LOCAL_RANK', '0'))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device('cuda', local_rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device('cpu')
    backend = 'nccl' if use_cuda else 'gloo'
    if world > 1:
        dist.init_process_group(backend=backend)
    else:
        print(json.dumps({'error': 'launch under torchrun with '
                          'WORLD_SIZE>1'}))
        return 1

    results = []
    for dname, dtype in DTYPES.items():
        if dtype == torch.bfloat16 and not use_cuda:
            continue
        for mb in SIZES_MB:
            n = int(mb * 1024 * 1024 / dtype.itemsize)
            t = torch.ones(n, dtype=dtype, device=device)
            iters = 50 if mb <= 8 else 20
            sec = bench_one(t, iters=iters, warmup=10, device=device)
            bytes_ = n * dtype.itemsize
            algbw = bytes_ / sec / 1e9
            busbw = algbw * 2 * (world - 1) / world
            row = {
                'collective': 'all_reduce', 'backend': backend,
                'world': world, 'dtype': dname, 'size_mb': mb,
                'us': round(sec * 1e6, 1), 'algbw_gbps': round(algbw, 2),
                'busbw_gbps': round(busbw, 2),
                'pct_of_link': round(100 * busbw / LINK_GBPS, 1),
            }
            results.append(row)
            if rank == 0:
                print(json.dumps(row), flush=True)

    if rank == 0:
        # pick the smallest size reaching >=70% of the peak measured busbw:
        # that's the knee — buckets below it waste bandwidth on latency
        gpu_rows = [r for r in results if r['dtype'] == 'fp32']
        if gpu_rows:
            peak = max(r['busbw_gbps'] for r in gpu_rows)
            knee = next((r for r in gpu_rows
                         if r['busbw_gbps'] >= 0.7 * peak), None)
            print(json.dumps({'summary': 'bucket-size knee',
                              'knee_mb': knee['size_mb'] if knee else None,
                              'peak_busbw_gbps': peak,
                              'note': 'set --bucket-cap-mb at or above the '
                                      'knee; smaller buckets only pay off '
                                      'if they hide under backward'}),
                  flush=True)
    dist.destroy_process_group()
    return 0


if __name__ == '__main__':
    sys.exit(main())
