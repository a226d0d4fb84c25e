"""Checkpoint format + resume round-trip + config sidecar restore."""

import os

import torch

from real_time_helmet_detection_amd.config import (build_parser,
                                                   get_arguments,
                                                   update_arguments_for_eval)
from real_time_helmet_detection_amd.models import StackedHourglass
from real_time_helmet_detection_amd.optim import get_optimizer
from real_time_helmet_detection_amd.loss import LossCalculator
from real_time_helmet_detection_amd import amp
from real_time_helmet_detection_amd.engine.checkpoint import (save_checkpoint,
                                                              load_checkpoint)


def _tiny_setup():
    net = StackedHourglass(num_stack=1, in_ch=8, out_ch=6)
    opt, sched = get_optimizer(net, lr=1e-3, lr_milestone=[5, 9],
                               lr_gamma=0.1)
    calc = LossCalculator()
    return net, opt, sched, calc


def test_checkpoint_dict_format(tmp_path):
    net, opt, sched, calc = _tiny_setup()
    scaler = amp.GradScaler()
    path = save_checkpoint(str(tmp_path), 3, net, opt, sched, scaler, calc)
    assert path.endswith('check_point_3.pth')
    ckpt = torch.load(path, map_location='cpu', weights_only=False)
    # mandated key set (reference train.py:76-82)
    assert set(ckpt.keys()) == {'epoch', 'state_dict', 'optimizer',
                                'scheduler', 'scaler', 'loss_log'}
    assert ckpt['epoch'] == 3


def test_resume_round_trip(tmp_path):
    torch.manual_seed(0)
    net, opt, sched, calc = _tiny_setup()
    x = torch.randn(2, 3, 64, 64)
    y = net(x)
    y.sum().backward()
    opt.step()
    sched.step()
    calc.log['total'].append(1.25)
    path = save_checkpoint(str(tmp_path), 1, net, opt, sched,
                           amp.GradScaler(), calc)

    net2, opt2, sched2, calc2 = _tiny_setup()
    load_checkpoint(path, net2, opt2, sched2, amp.GradScaler(), calc2)
    for (k1, v1), (k2, v2) in zip(net.state_dict().items(),
                                  net2.state_dict().items()):
        assert k1 == k2
        torch.testing.assert_close(v1, v2)
    assert sched2.last_epoch == sched.last_epoch
    assert calc2.log['total'] == [1.25]
    # optimizer state (Adam moments) survives
    s1 = opt.state_dict()['state']
    s2 = opt2.state_dict()['state']
    assert set(s1.keys()) == set(s2.keys())
    for k in s1:
        torch.testing.assert_close(s1[k]['exp_avg'], s2[k]['exp_avg'])


def test_ddp_unwrap_on_save(tmp_path):
    net, opt, sched, calc = _tiny_setup()

    class FakeWrap:
        def __init__(self, module):
            self.module = module
    path = save_checkpoint(str(tmp_path), 1, FakeWrap(net))
    ckpt = torch.load(path, map_location='cpu', weights_only=False)
    assert not any(k.startswith('module.') for k in ckpt['state_dict'])


def test_eval_arch_flag_restore(tmp_path, monkeypatch):
    save_path = str(tmp_path / 'w')
    args = get_arguments(['--train-flag', '--save-path', save_path,
                          '--num-stack', '3', '--hourglass-inch', '32',
                          '--activation', 'Mish'])
    assert os.path.isfile(os.path.join(save_path, 'argument.pickle'))
    assert os.path.isfile(os.path.join(save_path, 'argument.txt'))

    # eval-mode parse with default arch flags picks up the sidecar
    eval_args = get_arguments(['--save-path', str(tmp_path / 'e'),
                               '--model-load',
                               os.path.join(save_path, 'check_point_1.pth'),
                               '--imsize', '128'])
    assert eval_args.num_stack == 3
    assert eval_args.hourglass_inch == 32
    assert eval_args.activation == 'Mish'
    assert eval_args.imsize == 128  # non-arch flags untouched


def test_update_arguments_for_eval_dict_and_namespace():
    old = build_parser([])
    new = {'num_stack': 5, 'pool': 'Avg', 'imsize': 999}
    out = update_arguments_for_eval(old, new)
    assert out.num_stack == 5 and out.pool == 'Avg'
    assert out.imsize is None  # imsize is NOT an arch flag

    old2 = build_parser([])
    ns = build_parser(['--num-stack', '7'])
    out2 = update_arguments_for_eval(old2, ns)
    assert out2.num_stack == 7
