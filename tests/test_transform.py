"""Codec tests: box2hm <-> hm2box round trip (reference transform.py:112-131
generalized), peak properties, batched decode vs per-sample decode."""

import numpy as np
import pytest
import torch

from real_time_helmet_detection_amd.transform import (box2hm, hm2box,
                                                      draw_gaussian,
                                                      gaussian2D)
from real_time_helmet_detection_amd.ops import eager


@pytest.mark.parametrize('normalized', [False, True])
@pytest.mark.parametrize('box,label', [
    ([10, 20, 100, 200], 1),
    ([3, 3, 60, 50], 0),
    ([200, 300, 500, 510], 1),
])
def test_roundtrip(box, label, normalized):
    imsize = (512, 512)
    hm, off, wh, mask = box2hm([box], [label], imsize, normalized=normalized)
    assert hm.shape == (2, 128, 128)
    assert mask.sum() == 1.0

    boxes, clss, scores = hm2box(torch.from_numpy(hm), torch.from_numpy(off),
                                 torch.from_numpy(wh), topk=5, conf_th=0.5,
                                 normalized=normalized)
    assert len(boxes) >= 1
    assert int(clss[0]) == label
    assert float(scores[0]) == pytest.approx(1.0)
    np.testing.assert_allclose(boxes[0].numpy(), np.asarray(box, np.float32),
                               atol=1e-3)


def test_empty_boxes():
    hm, off, wh, mask = box2hm(None, None, (128, 128))
    assert hm.sum() == 0 and mask.sum() == 0


def test_gaussian_peak_is_one():
    g = gaussian2D((5, 5), sigma=2.0)
    assert g.shape == (11, 11)
    assert g[5, 5] == pytest.approx(1.0)


def test_draw_gaussian_max_splat():
    hm = np.zeros((32, 32), np.float32)
    draw_gaussian(hm, (10, 10), 6.0)
    before = hm.copy()
    draw_gaussian(hm, (12, 10), 6.0)  # overlapping second peak
    assert (hm >= before - 1e-7).all()  # max-splat never decreases
    assert hm[10, 10] == pytest.approx(1.0)
    assert hm[10, 12] == pytest.approx(1.0)


def test_batched_decode_matches_per_sample():
    torch.manual_seed(0)
    b, c, h, w = 3, 2, 32, 32
    hm = torch.rand(b, c, h, w)
    off = torch.rand(b, 2, h, w)
    wh = torch.rand(b, 2, h, w) * 10
    boxes, clss, scores = eager.batched_decode(hm, off, wh, scale_factor=4,
                                               topk=7, pool_size=3,
                                               normalized=False)
    assert boxes.shape == (b, 7, 4)
    for i in range(b):
        bi, ci, si = hm2box(hm[i], off[i], wh[i], scale_factor=4, topk=7,
                            conf_th=-1e9, normalized=False)
        torch.testing.assert_close(boxes[i], bi)
        torch.testing.assert_close(clss[i], ci)
        torch.testing.assert_close(scores[i], si)


@pytest.mark.parametrize('normalized', [False, True])
def test_batched_decode_normalized_paths(normalized):
    torch.manual_seed(1)
    hm = torch.rand(2, 2, 16, 16)
    off = torch.rand(2, 2, 16, 16)
    wh = torch.rand(2, 2, 16, 16)
    boxes, _, _ = eager.batched_decode(hm, off, wh, 4, 5, 3, normalized)
    for i in range(2):
        bi, _, _ = hm2box(hm[i], off[i], wh[i], 4, 5, -1e9, normalized)
        torch.testing.assert_close(boxes[i], bi)
