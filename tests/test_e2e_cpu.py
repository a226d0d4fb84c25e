"""Tiny-config end-to-end CPU slice: train -> checkpoint -> resume -> eval
(BASELINE.json config 1: 128px, batch 2, synthetic 2-class VOC boxes)."""

import os

import pytest
import torch

from real_time_helmet_detection_amd.config import build_parser
from real_time_helmet_detection_amd.engine.trainer import (distributed_worker,
                                                           load_network)
from real_time_helmet_detection_amd.engine.evaluator import \
    single_device_evaluate


def _train_args(tmp_path, extra=()):
    return build_parser([
        '--train-flag', '--synthetic', '--batch-size', '2',
        '--num-workers', '0', '--end-epoch', '1', '--synthetic-size', '4',
        '--multiscale', '64', '128', '64', '--print-interval', '100',
        '--hourglass-inch', '16', '--save-path',
        str(tmp_path) + os.sep, '--imsize', '128', *extra])


def test_train_eval_resume(tmp_path):
    os.makedirs(tmp_path / 'training_log', exist_ok=True)
    args = _train_args(tmp_path)
    distributed_worker(0, 1, args)
    ckpt = tmp_path / 'check_point_1.pth'
    assert ckpt.is_file()

    # resume for one more epoch from the checkpoint
    args2 = _train_args(tmp_path, extra=['--start-epoch', '1',
                                         '--end-epoch', '2',
                                         '--model-load', str(ckpt)])
    distributed_worker(0, 1, args2)
    assert (tmp_path / 'check_point_2.pth').is_file()

    # evaluation over the synthetic test split
    eval_args = build_parser([
        '--synthetic', '--batch-size', '2', '--num-workers', '0',
        '--synthetic-size', '4', '--imsize', '128', '--gpu-no', '-1',
        '--hourglass-inch', '16',
        '--save-path', str(tmp_path / 'eval') + os.sep,
        '--model-load', str(ckpt)])
    os.makedirs(tmp_path / 'eval', exist_ok=True)
    preds = single_device_evaluate(eval_args)
    assert len(preds) == 4
    assert os.path.isdir(tmp_path / 'eval' / 'txt')
    assert (tmp_path / 'eval' / 'prediction_results.pickle').is_file()


def test_grad_accumulation_equivalence(tmp_path):
    """--sub-divisions k over k micro-batches == one batch step (fp32)."""
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(0)
    netA = StackedHourglass(1, 8, 6)
    netB = StackedHourglass(1, 8, 6)
    netB.load_state_dict(netA.state_dict())
    # eval-mode BN: batch statistics would differ between the full batch and
    # the micro-batches, making exact equivalence impossible (true for the
    # reference's accumulation too); running-stat BN isolates the
    # accumulation plumbing.
    netA.eval()
    netB.eval()

    x = torch.randn(4, 3, 64, 64)

    # one full-batch backward (mean-style loss scaled like the trainer)
    netA(x).mean().backward()
    # two half-batches, each loss / sub_divisions
    (netB(x[:2]).mean() / 2).backward()
    (netB(x[2:]).mean() / 2).backward()

    for pa, pb in zip(netA.parameters(), netB.parameters()):
        torch.testing.assert_close(pa.grad, pb.grad, rtol=1e-4, atol=1e-5)


def test_loss_decreases_on_overfit():
    """Sanity: 30 steps on one fixed batch reduces the loss."""
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.loss import LossCalculator
    from real_time_helmet_detection_amd.engine.trainer import \
        compute_stack_losses
    from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor

    torch.manual_seed(0)
    ds = SyntheticVOC(transform=TestAugmentor(64), size=2, imsize=64, seed=3)
    img, hm, off, wh, mask, _ = ds.collate_fn([ds[0], ds[1]])

    net = StackedHourglass(1, 16, 6)
    calc = LossCalculator()
    opt = torch.optim.Adam(net.parameters(), lr=1e-3)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        out = net(img)
        total, _ = compute_stack_losses(out, calc, hm, off, wh, mask, 2,
                                        False)
        total.backward()
        opt.step()
        losses.append(total.item())
    assert losses[-1] < losses[0] * 0.5
