"""Flagship training-step benchmark (driver contract).

python bench.py --gpus N --steps K --warmup W
  (for N>1 the driver launches it under torch.distributed.run, one rank per
   GPU over RCCL; we read RANK/LOCAL_RANK/WORLD_SIZE from the env)

Measures the BASELINE.json headline: training imgs/sec of the hourglass
num-stack=1, in_ch=128, 512x512, bf16-AMP detector on synthetic
VOC2028-shaped data with random-init weights. One "step" = full training
iteration: forward, per-stack sigmoid+focal/L1 loss, backward, bucketed
RCCL all-reduce (N>1), Adam step. W untimed warmup steps, then EXACTLY K
timed steps bracketed by barrier + torch.cuda.synchronize on both sides;
elapsed = MAX over ranks; rank 0 prints one JSON line.

Weak scaling: per-GPU batch is fixed (16) as N grows.
"""

import argparse
import json
import os
import time

import torch


def build_args():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=20)
    p.add_argument('--warmup', type=int, default=5)
    p.add_argument('--batch-size', type=int, default=16,
                   help='per-GPU batch size (weak scaling)')
    p.add_argument('--imsize', type=int, default=512)
    p.add_argument('--num-stack', type=int, default=1)
    p.add_argument('--hourglass-inch', type=int, default=128)
    p.add_argument('--increase-ch', type=int, default=0)
    p.add_argument('--dtype', type=str, default='bf16',
                   choices=['bf16', 'fp32'])
    p.add_argument('--bucket-cap-mb', type=float, default=5.0)
    p.add_argument('--mode', type=str, default='train',
                   choices=['train', 'infer'],
                   help='train imgs/sec (headline) or inference FPS')
    p.add_argument('--engine', type=str, default='hip',
                   choices=['hip', 'torch'],
                   help='hip = in-tree gfx950 kernels; torch = eager '
                        'torch-ROCm ops (A/B baseline only)')
    p.add_argument('--fp8', action='store_true', default=False,
                   help='fp8 e4m3 MFMA inference (infer mode only)')
    p.add_argument('--no-train-graph', action='store_true', default=False,
                   help='disable hipGraph capture of the training step')
    p.add_argument('--graph', action='store_true', default=False,
                   help='hipGraph-captured inference (infer mode only)')
    p.add_argument('--channels-last', dest='channels_last',
                   action='store_true', default=True)
    p.add_argument('--no-channels-last', dest='channels_last',
                   action='store_false')
    return p.parse_args()


def make_synthetic_batches(batch_size, imsize, device, n_batches=2, seed=777):
    """Pre-encoded synthetic batches resident on the device (data loading is
    outside the timed region; data='synthetic' is declared in the output)."""
    from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
    ds = SyntheticVOC(transform=TestAugmentor(imsize), imsize=imsize,
                      size=batch_size * n_batches, seed=seed)
    batches = []
    for b in range(n_batches):
        items = [ds[b * batch_size + i] for i in range(batch_size)]
        img, hm, off, wh, mask, _ = ds.collate_fn(items)
        batches.append(tuple(t.to(device) for t in (img, hm, off, wh, mask)))
    return batches


def main():
    args = build_args()
    if args.engine == 'torch':
        os.environ['RTHD_EAGER_GPU'] = '1'

    env_launched = 'RANK' in os.environ and 'WORLD_SIZE' in os.environ
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1')) if env_launched else 1
    local_rank = int(os.environ.get('LOCAL_RANK', '0'))

    use_cuda = torch.cuda.is_available()
    # local_rank may exceed the visible device count when de-risking the
    # RCCL path with multiple ranks sharing one GPU (2 ranks on a 1-GPU
    # box); fold it into the available devices
    dev_idx = (local_rank % torch.cuda.device_count()) if use_cuda else 0
    device = torch.device('cuda', dev_idx) if use_cuda \
        else torch.device('cpu')
    if use_cuda:
        torch.cuda.set_device(device)

    if env_launched and world > 1:
        import torch.distributed as dist
        backend = 'nccl' if use_cuda else 'gloo'
        dist.init_process_group(backend=backend)
    else:
        dist = None

    torch.manual_seed(1234 + rank)

    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.loss import LossCalculator
    from real_time_helmet_detection_amd.engine.trainer import \
        compute_stack_losses
    from real_time_helmet_detection_amd import amp as rthd_amp

    num_cls = 2
    net = StackedHourglass(num_stack=args.num_stack,
                           in_ch=args.hourglass_inch,
                           out_ch=num_cls + 4,
                           increase_ch=args.increase_ch).to(device)
    if args.channels_last and use_cuda:
        net = net.to(memory_format=torch.channels_last)

    if world > 1:
        from real_time_helmet_detection_amd.parallel.ddp import \
            BucketedDataParallel
        model = BucketedDataParallel(net, bucket_cap_mb=args.bucket_cap_mb)
    else:
        model = net

    amp_on = args.dtype == 'bf16' and use_cuda
    batches = make_synthetic_batches(args.batch_size, args.imsize, device)
    if args.channels_last and use_cuda:
        batches = [(b[0].to(memory_format=torch.channels_last), b[1], b[2],
                    b[3], b[4]) for b in batches]

    calc = LossCalculator().to(device)
    is_bucketed = world > 1

    graph_active = False
    if args.mode == 'train':
        # N>1 captures the bucketed RCCL all-reduce into the graph too
        # (warmup establishes the comm clique first); any capture failure
        # falls back to eager stepping
        want_graph = use_cuda and not args.no_train_graph
        try:
            opt = torch.optim.Adam(net.parameters(), lr=5e-4, fused=True,
                                   capturable=want_graph)
        except (RuntimeError, TypeError, ValueError):
            opt = torch.optim.Adam(net.parameters(), lr=5e-4, foreach=True)
            want_graph = False

        def eager_step(i):
            img, hm, off, wh, mask = batches[i % len(batches)]
            with rthd_amp.autocast(enabled=amp_on):
                out = model(img)
            total, _ = compute_stack_losses(out, calc, hm, off, wh, mask,
                                            num_cls, False)
            total.backward()
            if is_bucketed:
                model.finish_backward()
            opt.step()
            opt.zero_grad(set_to_none=True)

        step = eager_step
        if want_graph:
            # hipGraph-captured training step (one graph per resident
            # synthetic batch): the ~400 kernel launches of a step become
            # one graph launch. Falls back to eager stepping on any
            # capture failure.
            try:
                for i in range(3):
                    eager_step(i)
                torch.cuda.synchronize()
                graphs = []
                for b in range(len(batches)):
                    g = torch.cuda.CUDAGraph()
                    # thread_local: the RCCL watchdog thread's HIP calls
                    # must not invalidate an N>1 capture
                    with torch.cuda.graph(
                            g, capture_error_mode='thread_local'):
                        eager_step(b)
                    graphs.append(g)
                calc._pending.clear()

                def step(i):
                    graphs[i % len(graphs)].replay()
                graph_active = True
            except Exception as e:
                print('train-graph capture failed (%s); eager stepping' % e)
                step = eager_step
    else:
        from real_time_helmet_detection_amd.engine.evaluator import (
            Prediction, GraphedPredictor)
        predictor = Prediction(net, topk=100, scale_factor=4, conf_th=0.0,
                               nms='nms', nms_th=0.5).to(device)
        predictor.eval()
        if args.graph and use_cuda:
            predictor = GraphedPredictor(predictor, batches[0][0],
                                         fp8=args.fp8)

            def step(i):
                img = batches[i % len(batches)][0]
                predictor(img)
        elif args.fp8 and use_cuda:
            def step(i):
                img = batches[i % len(batches)][0]
                with rthd_amp.autocast(enabled=amp_on), \
                        rthd_amp.fp8_autocast(True), torch.no_grad():
                    predictor(img)
        else:
            def step(i):
                img = batches[i % len(batches)][0]
                with rthd_amp.autocast(enabled=amp_on), torch.no_grad():
                    predictor(img)

    def barrier_sync():
        if dist is not None and world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        step(i)
    calc._pending.clear()
    barrier_sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist is not None and world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else 'cpu')
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if env_launched else (1 if use_cuda else 1)
    total_imgs = args.batch_size * n_gpus * args.steps
    value = total_imgs / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0:
        result = {
            'metric': ('train imgs/sec' if args.mode == 'train'
                       else 'inference FPS @%dx%d' % (args.imsize,
                                                      args.imsize)),
            'value': round(value, 2),
            'unit': 'imgs/sec',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(ms_per_step, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': ('fp8' if (args.fp8 and args.mode == 'infer')
                      else args.dtype) if use_cuda else 'fp32',
            'data': 'synthetic',
            'config': {
                'model': 'hourglass-%d-ch%d%s' % (
                    args.num_stack, args.hourglass_inch,
                    '-inc%d' % args.increase_ch if args.increase_ch else ''),
                'global_batch': args.batch_size * n_gpus,
                'imsize': args.imsize,
                'parallelism': 'dp%d' % n_gpus,
                'engine': args.engine,
                'fp8': args.fp8,
                'hipgraph': (graph_active if args.mode == 'train'
                             else args.graph),
            },
        }
        print(json.dumps(result))

    if dist is not None and world > 1:
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
