"""Training driver: per-rank worker, hot loop, model factory.

Behavioral parity with /root/reference/train.py:32-201 (worker setup, epoch
loop, per-stack deep-supervision loss with sigmoid applied OUTSIDE the net,
gradient accumulation via --sub-divisions, rank-0 logging + heatmap-blend
PNGs, per-epoch rank-0 checkpointing, load_network factory) with the MI355X
re-design:

- DDP is our BucketedDataParallel (RCCL over xGMI, buckets overlapped with
  backward); accumulation micro-steps run under ``no_sync`` instead of
  all-reducing every micro-batch like the reference did.
- AMP is bf16 (rthd.amp) — no loss scaling; losses are computed in fp32
  (log/pow near 0/1 need fp32; SURVEY.md §7 hard-parts #5).
- H2D copies are non_blocking on pinned batches; tensors run channels_last
  on GPU (the HIP kernels are NHWC-native).
- No per-iteration ``.item()`` syncs (LossCalculator defers the log flush).
"""

import os
import time
import contextlib
from collections import defaultdict

import torch
import torch.distributed as dist

from ..models import StackedHourglass
from ..loss import LossCalculator
from ..optim import get_optimizer
from ..data import load_dataset
from ..utils import AverageMeter, blend_heatmap
from .. import amp
from ..parallel.ddp import BucketedDataParallel
from ..parallel.launch import init_process_group_from_args
from .checkpoint import save_checkpoint

_COMM_DTYPES = {'fp32': torch.float32, 'bf16': torch.bfloat16}


def _resolve_device(args, local_rank):
    if torch.cuda.is_available() and args.gpu_no != [-1]:
        return torch.device('cuda', local_rank)
    return torch.device('cpu')


def distributed_worker(device, ngpus_per_node, args, env_launch=False):
    """One process per GPU (reference train.py:32-84)."""
    if env_launch:
        rank = int(os.environ['RANK'])
        world_size = int(os.environ['WORLD_SIZE'])
    else:
        rank = args.rank * ngpus_per_node + device
        world_size = args.world_size

    dev = _resolve_device(args, device)
    if dev.type == 'cuda':
        torch.cuda.set_device(dev)
    print('%s: worker rank %d/%d on %s' % (time.ctime(), rank, world_size,
                                           dev))

    batch_size = max(1, int(args.batch_size / ngpus_per_node))
    num_workers = int((args.num_workers + ngpus_per_node - 1)
                      / ngpus_per_node)

    if world_size > 1 and not dist.is_initialized():
        init_process_group_from_args(args, rank, world_size)

    network, optimizer, scheduler, loss_calculator = load_network(args, dev)
    if device == 0:
        base = network.module if hasattr(network, 'module') else network
        n_params = sum(p.numel() for p in base.parameters())
        print('%s: StackedHourglass(num_stack=%d, in_ch=%d, increase_ch=%d)'
              ': %.2fM parameters, engine=%s'
              % (time.ctime(), args.num_stack, args.hourglass_inch,
                 args.increase_ch, n_params / 1e6,
                 'hip' if dev.type == 'cuda' else 'cpu-eager'))

    dataset = load_dataset(args)
    sampler = torch.utils.data.distributed.DistributedSampler(dataset) \
        if world_size > 1 else None
    dataloader = torch.utils.data.DataLoader(
        dataset=dataset,
        batch_size=batch_size,
        shuffle=(sampler is None),
        num_workers=num_workers,
        pin_memory=(dev.type == 'cuda'),
        sampler=sampler,
        collate_fn=dataset.collate_fn,
        drop_last=False,
        persistent_workers=(num_workers > 0),
    )

    scaler = amp.GradScaler() if args.amp else None

    for epoch in range(args.start_epoch, args.end_epoch):
        if sampler is not None:
            sampler.set_epoch(epoch)
        train_step(dataloader, network, loss_calculator, optimizer,
                   scheduler, scaler, epoch, device, args)
        scheduler.step()
        if rank % ngpus_per_node == 0:
            save_checkpoint(args.save_path, epoch + 1, network, optimizer,
                            scheduler, scaler, loss_calculator)


def compute_stack_losses(outputs, loss_calculator, gt_heatmap, gt_offset,
                         gt_size, gt_mask, num_cls, normalized_coord):
    """Deep supervision: sum the loss over every stack's prediction.

    outputs: (B, S, num_cls+4, h, w) raw logits. Sigmoid on the heatmap
    channels (and on offset/size when normalized_coord) is applied on the
    way into the loss, outside the network (reference train.py:104-120), in
    fp32. On GPU the whole thing — sigmoid, upcast, all stacks — is ONE
    fused kernel pair (ops.hip.centernet_losses_logits).

    Returns (total_loss, last-stack heatmap LOGITS) — apply sigmoid before
    visualizing the heatmap.
    """
    from ..ops import _backend
    if outputs.is_cuda and not _backend.eager_gpu_override():
        from ..ops import hip
        losses = hip.centernet_losses_logits(
            outputs, gt_heatmap, gt_offset, gt_size, gt_mask,
            loss_calculator.focal_alpha, loss_calculator.focal_beta,
            normalized_coord)
        total = loss_calculator.accumulate_stack_losses(losses)
        return total, outputs[:, -1, :num_cls]

    total = None
    last_heatmap = None
    for output in outputs.split(1, dim=1):
        output = output.squeeze(1).float()
        hm_logits, pred_offset, pred_size = output.split(
            [num_cls, 2, 2], dim=1)
        pred_heatmap = torch.sigmoid(hm_logits)
        if normalized_coord:
            pred_offset = torch.sigmoid(pred_offset)
            pred_size = torch.sigmoid(pred_size)
        loss = loss_calculator(pred_heatmap, pred_offset, pred_size,
                               gt_heatmap, gt_offset, gt_size, gt_mask)
        total = loss if total is None else total + loss
        last_heatmap = hm_logits
    return total, last_heatmap


def _want_train_graph(args, device):
    """hipGraph capture of the full training step: GPU + HIP engine +
    no gradient accumulation (micro-steps change the per-iteration op
    sequence)."""
    device = torch.device(device) if not isinstance(device, torch.device) \
        else device
    world = dist.get_world_size() if dist.is_initialized() else 1
    # multi-rank multiscale would capture at DIFFERENT iterations per rank
    # (per-rank random batch sizes -> divergent shape keys): one rank
    # recording its all-reduce while another executes eagerly deadlocks
    # the collective sequence — run eagerly instead
    if world > 1 and getattr(args, 'multiscale_flag', False):
        return False
    return (getattr(args, 'train_graph', True)
            and device.type == 'cuda'
            and args.train_flag
            and getattr(args, 'sub_divisions', 1) == 1
            and os.environ.get('RTHD_EAGER_GPU') != '1')


def train_step(dataloader, network, loss_calculator, optimizer, scheduler,
               scaler, epoch, device, args):
    """One epoch (reference train.py:86-162)."""
    time_logger = defaultdict(AverageMeter)
    network.train()
    dev = next(network.parameters()).device
    use_cl = getattr(args, 'channels_last', True) and dev.type == 'cuda'
    n_batches = len(dataloader)

    is_bucketed = isinstance(network, BucketedDataParallel)

    graphed = getattr(network, '_rthd_graphed', None)
    if graphed is None and _want_train_graph(args, dev):
        from .graphed import GraphedTrainStep
        graphed = GraphedTrainStep(network, loss_calculator, optimizer,
                                   args.num_cls, args.normalized_coord,
                                   amp_on=(scaler is not None))
        network._rthd_graphed = graphed

    tictoc = time.time()
    for iteration, (image, gt_heatmap, gt_offset, gt_size, gt_mask,
                    gt_dict) in enumerate(dataloader, 1):
        time_logger['data'].update(time.time() - tictoc)

        step_now = (iteration % args.sub_divisions == 0) or \
            (iteration == n_batches)

        tictoc = time.time()
        image = image.to(dev, non_blocking=True)
        if use_cl:
            image = image.to(memory_format=torch.channels_last)
        gt_heatmap = gt_heatmap.to(dev, non_blocking=True)
        gt_offset = gt_offset.to(dev, non_blocking=True)
        gt_size = gt_size.to(dev, non_blocking=True)
        gt_mask = gt_mask.to(dev, non_blocking=True)

        if graphed is not None and graphed.enabled:
            # whole iteration (fwd + fused loss + bwd + allreduce + Adam)
            # as one hipGraph replay; falls through to eager on shape
            # overflow or capture failure
            hm_logits = graphed.step(image, gt_heatmap, gt_offset, gt_size,
                                     gt_mask)
            if hm_logits is not None:
                pred_heatmap = hm_logits
                time_logger['forward'].update(time.time() - tictoc)
                if (iteration % args.print_interval == 0) and (device == 0):
                    _print_train_log(time_logger, loss_calculator, epoch,
                                     iteration, n_batches, args, image,
                                     pred_heatmap, gt_heatmap)
                tictoc = time.time()
                continue

        with amp.autocast(enabled=scaler is not None):
            outputs = network(image)
        time_logger['forward'].update(time.time() - tictoc)

        tictoc = time.time()
        total_loss, pred_heatmap = compute_stack_losses(
            outputs, loss_calculator, gt_heatmap, gt_offset, gt_size,
            gt_mask, args.num_cls, args.normalized_coord)
        if args.sub_divisions > 1:
            total_loss = total_loss / args.sub_divisions
        time_logger['loss'].update(time.time() - tictoc)

        tictoc = time.time()
        sync_ctx = network.no_sync() if (is_bucketed and not step_now) \
            else contextlib.nullcontext()
        with sync_ctx:
            if scaler is not None:
                scaler.scale(total_loss).backward()
            else:
                total_loss.backward()
        if step_now:
            if is_bucketed:
                network.finish_backward()
            if scaler is not None:
                scaler.step(optimizer)
                scaler.update()
            else:
                optimizer.step()
            optimizer.zero_grad(set_to_none=True)
        time_logger['backward'].update(time.time() - tictoc)

        if (iteration % args.print_interval == 0) and (device == 0):
            _print_train_log(time_logger, loss_calculator, epoch, iteration,
                             n_batches, args, image, pred_heatmap,
                             gt_heatmap)

        tictoc = time.time()


def _print_train_log(time_logger, loss_calculator, epoch, iteration,
                     n_batches, args, image, pred_heatmap, gt_heatmap):
    """Rank-0 progress line + heatmap-blend PNGs (reference
    train.py:141-158). pred_heatmap holds LOGITS (sigmoid applied here,
    only when actually rendering)."""
    loss_log = loss_calculator.get_log()
    _log = '%s: Epoch [%2d/%2d]' % (time.ctime(), epoch, args.end_epoch)
    _log += ', Iteration [%4d/%4d]' % (iteration, n_batches)
    _log += ', Loss [%s]' % loss_log
    _log += ', Time(ms) [data: %6.2f' % (time_logger['data'].avg * 1e3)
    _log += ', forward: %6.2f' % (time_logger['forward'].avg * 1e3)
    _log += ', backward: %6.2f' % (time_logger['backward'].avg * 1e3)
    _log += ', loss: %6.2f]' % (time_logger['loss'].avg * 1e3)
    print(_log)

    log_dir = os.path.join(args.save_path, 'training_log')
    if os.path.isdir(log_dir):
        blend_heatmap(image[0], torch.sigmoid(pred_heatmap[0].float()),
                      args.pretrained).save(
            os.path.join(log_dir, 'training_pred.png'))
        blend_heatmap(image[0], gt_heatmap[0],
                      args.pretrained).save(
            os.path.join(log_dir, 'training_gt.png'))


def load_network(args, device):
    """Model/optim/loss factory + checkpoint restore
    (reference train.py:164-201)."""
    device = torch.device(device) if not isinstance(device, torch.device) \
        else device
    network = StackedHourglass(
        num_stack=args.num_stack,
        in_ch=args.hourglass_inch,
        out_ch=args.num_cls + 4,
        increase_ch=args.increase_ch,
        activation=args.activation,
        pool=args.pool,
        neck_activation=args.neck_activation,
        neck_pool=args.neck_pool).to(device)
    if getattr(args, 'channels_last', True) and device.type == 'cuda':
        network = network.to(memory_format=torch.channels_last)

    world_size = dist.get_world_size() if dist.is_initialized() else 1
    if world_size > 1 and args.train_flag:
        network = BucketedDataParallel(
            network,
            bucket_cap_mb=getattr(args, 'bucket_cap_mb', 5.0),
            comm_dtype=_COMM_DTYPES.get(getattr(args, 'comm_dtype', 'fp32'),
                                        torch.float32))

    optimizer, scheduler, loss_calculator = None, None, None
    if args.train_flag:
        optimizer, scheduler = get_optimizer(
            network=network, lr=args.lr, lr_milestone=args.lr_milestone,
            lr_gamma=args.lr_gamma, algo=args.optim,
            capturable=_want_train_graph(args, device))
        loss_calculator = LossCalculator(
            hm_weight=args.hm_weight,
            offset_weight=args.offset_weight,
            size_weight=args.size_weight,
            focal_alpha=args.focal_alpha,
            focal_beta=args.focal_beta).to(device)

    if args.model_load:
        ckpt = torch.load(args.model_load, map_location=device,
                          weights_only=False)
        (network.module if hasattr(network, 'module') else network
         ).load_state_dict(ckpt['state_dict'])
        print('%s: Weights are loaded from %s' % (time.ctime(),
                                                  args.model_load))
        if args.train_flag:
            if ckpt.get('optimizer') is not None:
                optimizer.load_state_dict(ckpt['optimizer'])
            if ckpt.get('loss_log') is not None:
                loss_calculator.load_loss_log(ckpt['loss_log'])
            if scheduler is not None and ckpt.get('scheduler') is not None:
                scheduler.load_state_dict(ckpt['scheduler'])
            # scaler state: our bf16 GradScaler is stateless (no loss
            # scaling needed on CDNA4); checkpoint.load_checkpoint restores
            # it when a stateful scaler is passed (reference never did —
            # SURVEY.md §5 bug note).

    return network, optimizer, scheduler, loss_calculator
