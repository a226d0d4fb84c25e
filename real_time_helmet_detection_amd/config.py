"""Flag system, seeding and argument persistence.

Behavior-compatible rebuild of the reference flag system
(/root/reference/config.py:11-179): same flag names/defaults/groups, the same
seeding of python/numpy/torch RNGs, the same ``argument.txt`` +
``argument.pickle`` dump next to checkpoints, and the same eval-time restore of
the 11 architecture flags from the checkpoint's sidecar pickle
(config.py:157-158, 171-179).

MI355X-specific additions live in their own group (``--engine``,
``--bucket-cap-mb``, ``--comm-dtype``, ``--channels-last``, ``--synthetic``):
they control the HIP kernel engine and the RCCL-over-xGMI gradient bucketing
and default to the MI355X-native path.
"""

import os
import argparse
import random

import numpy
import torch

from .utils import save_pickle, load_pickle

# The architecture flags that a checkpoint's sidecar pickle restores at eval
# time (reference config.py:171-179).
ARCH_FLAGS = [
    'scale_factor', 'num_cls', 'pretrained', 'normalized_coord',
    'num_stack', 'hourglass_inch', 'increase_ch', 'activation', 'pool',
    'neck_activation', 'neck_pool',
]


def make_parser():
    parser = argparse.ArgumentParser(
        description='MI355X-native real-time helmet detector')

    parser.add_argument('--gpu-no', type=int, nargs='+', default=[0],
            help='GPU ids to use, 0~N: GPU, -1: CPU')
    parser.add_argument('--random-seed', type=int, default=777,
            help='random seed for reproducible experiments')

    # train
    parser.add_argument('--train-flag', action='store_true', default=False,
            help='set this flag for training')
    parser.add_argument('--data', type=str, default=None,
            help='data path for training or evaluation')
    parser.add_argument('--batch-size', type=int, default=16,
            help='global batch size (split across GPUs)')
    parser.add_argument('--sub-divisions', type=int, default=1,
            help='optimize every N iterations for gradient accumulation. '
                 'NOTE: unlike the reference (which summed micro-batch '
                 'gradients), the accumulated loss is AVERAGED over the N '
                 'micro-batches (standard semantics); with N>1 the '
                 'effective step is 1/N of reference-tuned values')
    parser.add_argument('--start-epoch', type=int, default=0, help='start epoch')
    parser.add_argument('--end-epoch', type=int, default=100, help='end epoch')
    parser.add_argument('--num-workers', type=int, default=8,
            help='number of workers for data loading')

    # amp
    parser.add_argument('--amp', action='store_true', default=False,
            help='mixed precision flag (bf16 MFMA on MI355X; no loss scaling)')

    # ddp (distributed data parallel)
    parser.add_argument('--world-size', type=int, default=1,
            help='number of nodes for distributed data parallel')
    parser.add_argument('--rank', type=int, default=0,
            help='node rank for distributed data parallel')
    parser.add_argument('--dist-backend', type=str, default='nccl',
            help="torch.distributed backend ('nccl' IS RCCL on ROCm)")
    parser.add_argument('--dist-url', type=str,
            default='tcp://127.0.0.1:29500',
            help='rendezvous url for distributed data parallel')

    # evaluation and demo
    parser.add_argument('--imsize', type=int, default=None,
            help='evaluation/demo image resize (imsize x imsize)')
    parser.add_argument('--topk', type=int, default=100,
            help='extract topk peak predictions')
    parser.add_argument('--conf-th', type=float, default=0.0,
            help='confidence threshold')
    parser.add_argument('--nms-th', type=float, default=0.5,
            help='nms threshold')
    parser.add_argument('--pool-size', type=int, default=3,
            help='pool size used to find peak values in the heatmap')
    parser.add_argument('--model-load', type=str, default=None,
            help='check_point path')
    parser.add_argument('--nms', type=str, default='nms',
            help='select nms algorithm (nms | soft-nms)')
    parser.add_argument('--fontsize', type=int, default=10,
            help='fontsize for demo, 0: do not write score/class in the image')

    # augmentation
    parser.add_argument('--crop-percent', type=float, nargs='+',
            default=[0.0, 0.1], help='range(min, max) crop fraction')
    parser.add_argument('--color-multiply', type=float, nargs='+',
            default=[1.2, 1.5], help='range(min, max) brightness multiply')
    parser.add_argument('--translate-percent', type=float, default=0.1,
            help='translation ratio')
    parser.add_argument('--affine-scale', type=float, nargs='+',
            default=[0.5, 1.5], help='range(min, max) affine scaling')
    parser.add_argument('--multiscale_flag', action='store_true', default=False,
            help='per-iteration random input resolution')
    parser.add_argument('--multiscale', type=int, nargs='+',
            default=[320, 512, 64],
            help='[min, max, step]; without multiscale_flag train at max')

    # loss
    parser.add_argument('--hm-weight', type=float, default=1.0,
            help='heatmap loss weight')
    parser.add_argument('--offset-weight', type=float, default=1.0,
            help='offset loss weight')
    parser.add_argument('--size-weight', type=float, default=0.1,
            help='size (wh) loss weight')
    parser.add_argument('--focal-alpha', type=float, default=2.0,
            help='alpha for the heatmap focal loss')
    parser.add_argument('--focal-beta', type=float, default=4.0,
            help='beta for the heatmap focal loss')

    # network
    parser.add_argument('--scale_factor', type=int, default=4,
            help='downsampling scale from image to heatmap')
    parser.add_argument('--num-cls', type=int, default=2,
            help='number of classes')
    parser.add_argument('--pretrained', type=str, default='imagenet',
            help='input normalization scheme (scratch | imagenet)')
    parser.add_argument('--normalized-coord', action='store_true', default=False,
            help='predict normalized (relative) offset and size')
    # backbone - hourglass
    parser.add_argument('--num-stack', type=int, default=1,
            help='number of stacks in the hourglass network')
    parser.add_argument('--hourglass-inch', type=int, default=128,
            help='number of channels in the hourglass network')
    parser.add_argument('--increase-ch', type=int, default=0,
            help='channel increase per hourglass depth level')
    parser.add_argument('--activation', type=str, default='ReLU',
            help='activation function')
    parser.add_argument('--pool', type=str, default='Max',
            help='pooling function')
    # neck
    parser.add_argument('--neck-activation', type=str, default='ReLU',
            help='neck activation function')
    parser.add_argument('--neck-pool', type=str, default='None',
            help='neck pooling function (None | SPP)')

    # optimization
    parser.add_argument('--lr', type=float, default=5e-4, help='learning rate')
    parser.add_argument('--optim', type=str, default='Adam',
            help='optimization algorithm')
    parser.add_argument('--lr-milestone', type=int, nargs='+', default=[50, 90],
            help='epochs at which lr is scaled by lr-gamma')
    parser.add_argument('--lr-gamma', type=float, default=0.1,
            help='lr scale factor at milestones')

    # log
    parser.add_argument('--print-interval', type=int, default=100,
            help='print logs every N iterations')
    parser.add_argument('--save-path', type=str, default='./WEIGHTS/',
            help='path to save results')

    # --- MI355X-native group (not in the reference CLI) ---
    parser.add_argument('--engine', type=str, default='auto',
            help='compute engine: hip (CDNA4 kernels, GPU only) | torch | auto')
    parser.add_argument('--bucket-cap-mb', type=float, default=5.0,
            help='gradient all-reduce bucket size in MiB '
                 '(sized for 7 xGMI p2p links per GPU, not NVSwitch)')
    parser.add_argument('--comm-dtype', type=str, default='fp32',
            help='gradient all-reduce dtype (fp32 | bf16)')
    parser.add_argument('--channels-last', action='store_true', default=True,
            help='NHWC tensor layout (the HIP kernels are NHWC-native)')
    parser.add_argument('--no-channels-last', dest='channels_last',
            action='store_false')
    parser.add_argument('--train-graph', dest='train_graph',
            action='store_true', default=True,
            help='hipGraph-capture the training step (one graph launch '
                 'per iteration instead of ~150 kernel launches); '
                 'auto-disabled for --sub-divisions > 1 and on any '
                 'capture failure')
    parser.add_argument('--no-train-graph', dest='train_graph',
            action='store_false')
    parser.add_argument('--synthetic', action='store_true', default=False,
            help='use synthetic VOC2028-shaped data (no dataset on disk)')
    parser.add_argument('--synthetic-size', type=int, default=512,
            help='number of synthetic samples per epoch')
    return parser


def build_parser(argv=None):
    """Parse argv (public interface parity: reference config.py:11-136)."""
    return make_parser().parse_args(argv)


def seed_everything(seed):
    random.seed(seed)
    numpy.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
        torch.cuda.manual_seed_all(seed)


def get_arguments(argv=None):
    """Parse + seed + persist args (reference config.py:139-169).

    Train mode: creates save dirs and dumps ``argument.txt``/``argument.pickle``.
    Eval mode: restores the architecture flags from the ``argument.pickle``
    sitting next to ``--model-load`` so checkpoints are self-describing.
    """
    args = build_parser(argv)
    seed_everything(args.random_seed)

    os.makedirs(args.save_path, exist_ok=True)
    if args.train_flag:
        os.makedirs(os.path.join(args.save_path, 'training_log'), exist_ok=True)
    elif args.model_load is not None:
        sidecar = os.path.join(os.path.dirname(args.model_load),
                               'argument.pickle')
        if os.path.isfile(sidecar):
            args = update_arguments_for_eval(args, load_pickle(sidecar))

    # Restrict visible devices (HIP honors CUDA_VISIBLE_DEVICES on ROCm).
    if args.gpu_no != [-1]:
        os.environ.setdefault('CUDA_DEVICE_ORDER', 'PCI_BUS_ID')
        os.environ['CUDA_VISIBLE_DEVICES'] = ','.join(map(str, args.gpu_no))

    if getattr(args, 'engine', 'auto') == 'torch':
        # explicit eager-torch engine on GPU (A/B measurement only)
        os.environ['RTHD_EAGER_GPU'] = '1'

    with open(os.path.join(args.save_path, 'argument.txt'), 'w') as f:
        for key, value in sorted(vars(args).items()):
            f.write('%s: %s' % (key, value) + '\n')
    save_pickle(os.path.join(args.save_path, 'argument.pickle'), vars(args))
    return args


def update_arguments_for_eval(old, new):
    """Overwrite architecture flags from a loaded sidecar (config.py:171-179).

    ``new`` may be a Namespace or a plain dict (we persist dicts; a dict also
    survives module renames across versions).
    """
    src = new if isinstance(new, dict) else vars(new)
    for target in ARCH_FLAGS:
        if target in src:
            setattr(old, target, src[target])
    return old
