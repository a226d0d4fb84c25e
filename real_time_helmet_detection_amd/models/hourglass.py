"""Stacked-hourglass CenterNet backbone + heads, MI355X-native.

Architecture parity with /root/reference/hourglass.py:6-237 (block-by-block
census in SURVEY.md §2.2): PreLayer (7x7 s2 -> residuals, net /4), recursive
depth-4 Hourglass with Max/Avg/Conv/SPP/None pooling, Neck (optional SPP +
1x1 conv + residual), 1x1 linear Head per stack, inter-stack merge convs, and
deep supervision via ``torch.stack(predictions, dim=1)`` ->
``(B, num_stack, num_cls+4, H/4, W/4)``. Default config = 4.98 M params
(verified by tests/test_model.py).

Module attribute names match the reference state_dict layout
(pre_layer.layers.0.convolution.weight, hourglass_lst.0.up1.conv1...) so
reference-trained checkpoints load directly.

All hot ops route through ops.functional — on GPU that is the hand-written
gfx950 kernel set (fused conv+BN+act, pool, fused upsample+add); on CPU,
eager torch. Forward semantics are identical.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as F2

# Activations the fused HIP conv epilogue implements directly. Everything
# else goes through the module path after the conv+BN kernel.
FUSIBLE_ACTS = {'ReLU', 'Linear', 'LReLU'}


class Mish(nn.Module):
    def forward(self, x):
        return x * torch.tanh(F.softplus(x))


class Activation(nn.Module):
    """Name -> activation dispatch (reference hourglass.py:14-43)."""

    def __init__(self, activation: str):
        super().__init__()
        self.name = activation
        if activation == 'ReLU':
            self.activation = nn.ReLU()
        elif activation == 'LReLU':
            self.activation = nn.LeakyReLU()
        elif activation == 'PReLU':
            self.activation = nn.PReLU()
        elif activation == 'Linear':
            self.activation = nn.Identity()
        elif activation == 'Mish':
            self.activation = Mish()
        elif activation == 'Sigmoid':
            self.activation = nn.Sigmoid()
        elif activation == 'CELU':
            self.activation = nn.CELU()
        else:
            raise NotImplementedError(f'Not expected activation: {activation}')

    def forward(self, x):
        return self.activation(x)


class SPP(nn.Module):
    """YOLOv4-style SPP: 1x1 halve -> maxpools k in {5,9,13} s1 -> concat ->
    1x1 restore (reference hourglass.py:46-65)."""

    def __init__(self, ch=128, kernel_sizes=(5, 9, 13), stride=1):
        super().__init__()
        _ch = ch // 2
        self.conv1 = nn.Conv2d(ch, _ch, 1, 1, bias=False)
        self.conv2 = nn.Conv2d(_ch * (len(kernel_sizes) + 1), ch, 1, 1,
                               bias=False)
        self.pooling_layers = nn.ModuleList(
            [nn.MaxPool2d(k, stride, (k - 1) // 2) for k in kernel_sizes])
        self.kernel_sizes = list(kernel_sizes)

    def forward(self, x):
        x = F2.conv_bn_act(x, self.conv1, None, 'Linear', None, self.training)
        branches = [x] + [F2.maxpool_same(x, k) for k in self.kernel_sizes]
        y = torch.cat(branches, dim=1)
        return F2.conv_bn_act(y, self.conv2, None, 'Linear', None,
                              self.training)


class Pool(nn.Module):
    """Max/Avg/Conv/SPP/None pool dispatch (reference hourglass.py:68-91).

    Note the reference quirk kept on purpose: pool='SPP' does NOT downsample
    and its *declared* channel contract quadruples (callers pass in_ch*4 to
    the next block) even though SPP itself returns ch channels — we preserve
    the reference's structure exactly so configs behave identically.
    """

    def __init__(self, channel: int, pool: str):
        super().__init__()
        self.kind = pool
        if pool == 'Max':
            self.pool = nn.MaxPool2d(2, 2)
        elif pool == 'Avg':
            self.pool = nn.AvgPool2d(2, 2)
        elif pool == 'Conv':
            self.pool = nn.Conv2d(channel, channel, kernel_size=2, stride=2)
        elif pool == 'SPP':
            self.pool = SPP(channel)
        elif pool == 'None':
            self.pool = nn.Identity()
        else:
            raise NotImplementedError(f'Not expected pool: {pool}')

    def forward(self, x):
        if self.kind == 'Max':
            return F2.maxpool2x2(x)
        if self.kind == 'Avg':
            return F2.avgpool2x2(x)
        if self.kind == 'Conv':
            return F2.conv_bn_act(x, self.pool, None, 'Linear', None,
                                  self.training)
        return self.pool(x)


class Convolution(nn.Module):
    """conv(k, same-pad) -> optional BN -> activation
    (reference hourglass.py:94-108). Runs as ONE fused HIP kernel on GPU."""

    def __init__(self, in_ch, out_ch, kernel_size=3, stride=1, bias=True,
                 bn=False, activation='ReLU'):
        super().__init__()
        self.activation = Activation(activation)
        self.convolution = nn.Conv2d(in_ch, out_ch, kernel_size, stride,
                                     padding=(kernel_size - 1) // 2, bias=bias)
        self.bn = nn.BatchNorm2d(out_ch, affine=True,
                                 track_running_stats=True) if bn \
            else nn.Identity()

    def forward(self, x, skip=None, act_override=None):
        bn = self.bn if isinstance(self.bn, nn.BatchNorm2d) else None
        name = act_override or self.activation.name
        if name in FUSIBLE_ACTS:
            return F2.conv_bn_act(x, self.convolution, bn, name, None,
                                  self.training, skip=skip)
        y = F2.conv_bn_act(x, self.convolution, bn, 'Linear', None,
                           self.training, skip=skip)
        return self.activation(y)


class Residual(nn.Module):
    """conv3x3(BN,act) -> conv3x3(BN,linear) + skip -> act
    (reference hourglass.py:111-127)."""

    def __init__(self, in_ch, out_ch, kernel_size=3, stride=1,
                 activation='ReLU'):
        super().__init__()
        self.activation = Activation(activation)
        self.conv1 = Convolution(in_ch, out_ch, kernel_size, stride,
                                 bias=False, bn=True, activation=activation)
        self.conv2 = Convolution(out_ch, out_ch, kernel_size, stride,
                                 bias=False, bn=True, activation='Linear')
        if in_ch != out_ch:
            self.skip = Convolution(in_ch, out_ch, kernel_size=1,
                                    stride=stride, bias=False, bn=True,
                                    activation='Linear')
        else:
            self.skip = nn.Identity()

    def forward(self, x):
        y1 = self.conv1(x)
        s = x if isinstance(self.skip, nn.Identity) else self.skip(x)
        name = self.activation.name
        if name in FUSIBLE_ACTS:
            # fuse skip-add + outer activation into conv2's epilogue
            return self.conv2(y1, skip=s, act_override=name)
        return self.activation(self.conv2(y1) + s)


class Hourglass(nn.Module):
    """Recursive hourglass of depth num_layer (reference hourglass.py:130-156).

    up1 = Residual at this resolution; down path pools then
    low1(in->mid) -> low2(recurse|Residual) -> low3(mid->in) -> nearest 2x
    upsample; output = up1 + up2 (fused upsample+add on GPU).
    """

    def __init__(self, num_layer, in_ch, increase_ch=0, activation='ReLU',
                 pool='Max'):
        super().__init__()
        mid_ch = in_ch + increase_ch
        self.up1 = Residual(in_ch, in_ch, activation=activation)
        self.pool1 = Pool(in_ch, pool=pool)
        _in_ch = in_ch * 4 if pool == 'SPP' else in_ch
        self.low1 = Residual(_in_ch, mid_ch, activation=activation)
        if num_layer > 1:
            self.low2 = Hourglass(num_layer - 1, mid_ch, increase_ch,
                                  activation=activation, pool=pool)
        else:
            self.low2 = Residual(mid_ch, mid_ch, activation=activation)
        self.low3 = Residual(mid_ch, in_ch, activation=activation)
        self.up2 = nn.Upsample(scale_factor=2, mode='nearest')

    def forward(self, x):
        up1 = self.up1(x)
        low = self.low3(self.low2(self.low1(self.pool1(x))))
        return F2.upsample2x_add(low, up1)


class PreLayer(nn.Module):
    """Stem: 7x7 s2 conv(3->64,BN) -> Residual(64->mid) -> Pool ->
    Residual -> Residual(mid->out); net /4 resolution
    (reference hourglass.py:159-173)."""

    def __init__(self, in_ch=3, mid_ch=128, out_ch=5, activation='ReLU',
                 pool='Max'):
        super().__init__()
        layers = [
            Convolution(in_ch=in_ch, out_ch=64, kernel_size=7, stride=2,
                        bias=True, bn=True, activation=activation),
            Residual(in_ch=64, out_ch=mid_ch),
            Pool(channel=mid_ch, pool=pool),
        ]
        _mid_ch = mid_ch * 4 if pool == 'SPP' else mid_ch
        layers.append(Residual(in_ch=_mid_ch, out_ch=mid_ch))
        layers.append(Residual(in_ch=mid_ch, out_ch=out_ch))
        self.layers = nn.Sequential(*layers)

    def forward(self, x):
        return self.layers(x)


class Neck(nn.Module):
    """Pool(None|SPP) -> 1x1 conv-BN-act -> Residual
    (reference hourglass.py:176-186)."""

    def __init__(self, ch=128, activation='ReLU', pool='None'):
        super().__init__()
        self.layers = nn.Sequential(
            Pool(ch, pool),
            Convolution(in_ch=ch, out_ch=ch, kernel_size=1, bn=True,
                        activation=activation),
            Residual(ch, ch),
        )

    def forward(self, x):
        return self.layers(x)


class Head(nn.Module):
    """Single 1x1 linear conv with bias (reference hourglass.py:189-195)."""

    def __init__(self, in_ch, out_ch, kernel_size=1, stride=1, bias=True,
                 bn=False, activation='Linear'):
        super().__init__()
        self.layer = Convolution(in_ch=in_ch, out_ch=out_ch,
                                 kernel_size=kernel_size, stride=stride,
                                 bias=bias, bn=bn, activation=activation)

    def forward(self, x):
        return self.layer(x)


class StackedHourglass(nn.Module):
    """Full detector backbone (reference hourglass.py:198-237).

    forward: (B,3,H,W) -> (B, num_stack, out_ch, H/4, W/4); channel order
    [heatmap(num_cls), offset(2), wh(2)]; sigmoid is applied OUTSIDE the
    network (train/eval drivers) so the exported graph keeps linear heads.
    """

    def __init__(self, num_stack, in_ch, out_ch, increase_ch=0,
                 activation='ReLU', pool='Max', neck_activation='ReLU',
                 neck_pool='None'):
        super().__init__()
        self.pre_layer = PreLayer(in_ch=3, mid_ch=128, out_ch=in_ch,
                                  activation=activation, pool=pool)
        self.hourglass_lst = nn.ModuleList([
            Hourglass(num_layer=4, in_ch=in_ch, increase_ch=increase_ch,
                      activation=activation, pool=pool)
            for _ in range(num_stack)])
        self.neck_lst = nn.ModuleList([
            Neck(in_ch, neck_activation, neck_pool) for _ in range(num_stack)])
        self.head_lst = nn.ModuleList([
            Head(in_ch=in_ch, out_ch=out_ch) for _ in range(num_stack)])
        self.merge_feature = nn.ModuleList([
            Convolution(in_ch=in_ch, out_ch=in_ch, kernel_size=1, stride=1,
                        bias=True, bn=False, activation='Linear')
            for _ in range(num_stack - 1)])
        self.merge_prediction = nn.ModuleList([
            Convolution(in_ch=out_ch, out_ch=in_ch, kernel_size=1, stride=1,
                        bias=True, bn=False, activation='Linear')
            for _ in range(num_stack - 1)])
        self.num_stack = num_stack

    def forward(self, x):
        x = self.pre_layer(x)
        predictions = []
        for i in range(self.num_stack):
            hg = self.hourglass_lst[i](x)
            feature = self.neck_lst[i](hg)
            prediction = self.head_lst[i](feature)
            predictions.append(prediction)
            if i < self.num_stack - 1:
                # x + merge_feature(feature) + merge_prediction(prediction)
                # (reference hourglass.py:234-235) with both adds fused
                # into the 1x1 convs' epilogues
                mf = self.merge_feature[i](feature, skip=x)
                x = self.merge_prediction[i](prediction, skip=mf)
        return torch.stack(predictions, dim=1)


def build_model(args):
    """Model factory from parsed args (out_ch = num_cls + 4)."""
    return StackedHourglass(
        num_stack=args.num_stack,
        in_ch=args.hourglass_inch,
        out_ch=args.num_cls + 4,
        increase_ch=args.increase_ch,
        activation=args.activation,
        pool=args.pool,
        neck_activation=args.neck_activation,
        neck_pool=args.neck_pool,
    )
