"""Loss tests: formula parity against an independent direct implementation
(the reference math) + semantic properties."""

import math

import pytest
import torch

from real_time_helmet_detection_amd.loss import (LossCalculator, FocalLoss,
                                                 NormedL1Loss)
from real_time_helmet_detection_amd.ops import eager


def _ref_focal(pred, gt, mask, alpha, beta, eps=1e-7):
    """Direct transcription of the reference formula for oracle use."""
    neg_inds = torch.ones_like(mask) - mask
    neg_weights = torch.pow(1 - gt, beta)
    pos_loss = torch.log(pred + eps) * torch.pow(1 - pred, alpha) * mask
    neg_loss = torch.log(1 - pred + eps) * torch.pow(pred, alpha) \
        * neg_weights * neg_inds
    pos_loss = pos_loss.sum(dim=[1, 2, 3]).mean()
    neg_loss = neg_loss.sum(dim=[1, 2, 3]).mean()
    num_pos = mask.sum().clamp(1, 1e30)
    return -(pos_loss + neg_loss) / num_pos


def _ref_l1(pred, gt, mask):
    loss = torch.abs(pred * mask - gt * mask)
    loss = torch.sum(loss, dim=[1, 2, 3]).mean()
    return loss / mask.sum().clamp(1, 1e30)


def _random_inputs(b=2, c=2, h=16, w=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    phm = torch.rand(b, c, h, w, generator=g).clamp(1e-4, 1 - 1e-4)
    ghm = torch.rand(b, c, h, w, generator=g)
    mask = (torch.rand(b, 1, h, w, generator=g) > 0.95).float()
    poff = torch.randn(b, 2, h, w, generator=g)
    goff = torch.rand(b, 2, h, w, generator=g)
    psize = torch.randn(b, 2, h, w, generator=g) * 5
    gsize = torch.rand(b, 2, h, w, generator=g) * 10
    # gt heatmap is 1.0 at masked centers (box2hm contract)
    ghm = torch.maximum(ghm, mask.expand_as(ghm) * 0.999)
    return phm, poff, psize, ghm, goff, gsize, mask


def test_focal_matches_reference_formula():
    phm, _, _, ghm, _, _, mask = _random_inputs()
    got = FocalLoss(2.0, 4.0)(phm, ghm, mask)
    want = _ref_focal(phm, ghm, mask, 2.0, 4.0)
    torch.testing.assert_close(got, want)


def test_l1_matches_reference_formula():
    _, poff, _, _, goff, _, mask = _random_inputs()
    got = NormedL1Loss()(poff, goff, mask)
    want = _ref_l1(poff, goff, mask)
    torch.testing.assert_close(got, want)


def test_centernet_losses_bundle():
    phm, poff, psize, ghm, goff, gsize, mask = _random_inputs(seed=3)
    hm, off, size = eager.centernet_losses(phm, poff, psize, ghm, goff,
                                           gsize, mask, 2.0, 4.0)
    torch.testing.assert_close(hm, _ref_focal(phm, ghm, mask, 2.0, 4.0))
    torch.testing.assert_close(off, _ref_l1(poff, goff, mask))
    torch.testing.assert_close(size, _ref_l1(psize, gsize, mask))


def test_perfect_prediction_low_loss():
    _, _, _, ghm, goff, gsize, mask = _random_inputs(seed=5)
    ghm = mask.expand(-1, 2, -1, -1).clone()  # exact 0/1 heatmap
    phm = ghm.clamp(1e-6, 1 - 1e-6)
    hm, off, size = eager.centernet_losses(
        phm, goff, gsize, ghm, goff, gsize, mask, 2.0, 4.0)
    assert hm.item() < 1e-3
    assert off.item() == 0.0
    assert size.item() == 0.0


def test_no_positives_does_not_nan():
    phm, poff, psize, ghm, goff, gsize, _ = _random_inputs(seed=7)
    mask = torch.zeros(2, 1, 16, 16)
    hm, off, size = eager.centernet_losses(phm, poff, psize, ghm, goff,
                                           gsize, mask, 2.0, 4.0)
    assert math.isfinite(hm.item())
    assert off.item() == 0.0


def test_loss_calculator_log_and_weights():
    phm, poff, psize, ghm, goff, gsize, mask = _random_inputs(seed=9)
    calc = LossCalculator(hm_weight=1.0, offset_weight=1.0, size_weight=0.1,
                          focal_alpha=2.0, focal_beta=4.0)
    total = calc(phm, poff, psize, ghm, goff, gsize, mask)
    hm = _ref_focal(phm, ghm, mask, 2.0, 4.0)
    off = _ref_l1(poff, goff, mask)
    size = _ref_l1(psize, gsize, mask)
    torch.testing.assert_close(total, hm + off + 0.1 * size)
    # deferred log flush
    assert calc.log['total'] == []
    log_str = calc.get_log()
    assert len(calc.log['total']) == 1
    assert 'total' in log_str
    # grad flows
    phm2 = phm.clone().requires_grad_(True)
    calc(phm2, poff, psize, ghm, goff, gsize, mask).backward()
    assert phm2.grad is not None and torch.isfinite(phm2.grad).all()
