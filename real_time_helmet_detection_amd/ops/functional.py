"""Functional forms used by the model modules.

This is the seam where the StackedHourglass blocks meet the compute engines:
CPU tensors run plain torch eager; CUDA tensors run the gfx950 HIP kernels
(fused conv+BN+act implicit-GEMM, pooling, nearest upsample, fused
residual-add+act). Autograd flows through torch for eager and through custom
autograd.Functions for the HIP kernels — same signatures either way, so the
model code never branches.

amp note: when ``rthd.amp`` autocast is active and the tensor is on GPU the
conv paths run bf16 MFMA with fp32 accumulate (no loss scaling needed).
"""

import torch
import torch.nn.functional as F

from . import _backend
from .. import amp as _amp


def _hip(x):
    return x.is_cuda and not _backend.eager_gpu_override()


def _maybe_bf16(x):
    if _amp.is_autocast_enabled() and x.is_cuda and x.dtype == torch.float32:
        return x.to(torch.bfloat16)
    return x


def bn_momentum(bn, training):
    """nn.BatchNorm2d bookkeeping the functional call skips: increment
    num_batches_tracked on training forwards and resolve momentum=None to
    the cumulative-moving-average factor 1/num_batches_tracked."""
    if training and bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    if bn.momentum is None:
        n = (int(bn.num_batches_tracked.item())
             if bn.num_batches_tracked is not None else 1)
        return 1.0 / max(n, 1)
    return bn.momentum


def _named_act(y, act):
    """Apply the fused-epilogue activation names on the eager path."""
    if act == 'ReLU':
        return F.relu(y)
    if act == 'LReLU':
        return F.leaky_relu(y, 0.01)
    if act == 'Linear' or act is None:
        return y
    raise ValueError(f'non-fusible activation {act!r} must go through '
                     'act_module')


def conv_bn_act(x, conv, bn=None, act='Linear', act_module=None,
                training=False, skip=None):
    """conv -> (BN) -> (+skip) -> activation, fused on the HIP path.

    conv: nn.Conv2d holding weight/bias; bn: nn.BatchNorm2d or None;
    act: activation name; act_module: the nn.Module for parametric /
    non-fusible activations (PReLU etc. — applied after a Linear epilogue);
    skip: optional residual tensor added BEFORE the activation (the
    Residual tail fuses its add+act into this epilogue).
    """
    if _hip(x):
        from . import hip
        return hip.conv_bn_act(x, conv, bn, act, act_module, training, skip)

    y = F.conv2d(x, conv.weight, conv.bias, stride=conv.stride,
                 padding=conv.padding)
    if bn is not None:
        y = F.batch_norm(y, bn.running_mean, bn.running_var, bn.weight,
                         bn.bias, training=training,
                         momentum=bn_momentum(bn, training), eps=bn.eps)
    if skip is not None:
        y = y + skip
    if act_module is not None:
        return act_module(y)
    return _named_act(y, act)


def add_act(a, b, act='Linear', act_module=None):
    """Residual add followed by activation (fused elementwise on HIP)."""
    if _hip(a):
        from . import hip
        return hip.add_act(a, b, act, act_module)
    y = a + b
    if act_module is not None:
        return act_module(y)
    return _named_act(y, act)


def maxpool2x2(x):
    if _hip(x):
        from . import hip
        return hip.maxpool2x2(x)
    return F.max_pool2d(x, 2, 2)


def avgpool2x2(x):
    if _hip(x):
        from . import hip
        return hip.avgpool2x2(x)
    return F.avg_pool2d(x, 2, 2)


def maxpool_same(x, kernel):
    if _hip(x):
        from . import hip
        return hip.maxpool_same(x, kernel)
    return F.max_pool2d(x, kernel, stride=1, padding=kernel // 2)


def upsample2x_add(x, skip=None):
    """Nearest 2x upsample, optionally fused with the hourglass skip add."""
    if _hip(x):
        from . import hip
        return hip.upsample2x_add(x, skip)
    y = F.interpolate(x, scale_factor=2, mode='nearest')
    if skip is not None:
        y = y + skip
    return y
