"""mAP metric + NMS + evaluator plumbing tests."""

import numpy as np
import pytest
import torch

from real_time_helmet_detection_amd.engine.metrics import (voc_map,
                                                           average_precision)
from real_time_helmet_detection_amd.engine.evaluator import (
    resize_box_to_original_scale, write_detection_txt)
from real_time_helmet_detection_amd.ops import eager


def test_perfect_predictions_map_1():
    gt = {
        'a.jpg': (np.array([[10., 10., 50., 50.], [60., 60., 90., 90.]]),
                  np.array([0, 1])),
        'b.jpg': (np.array([[5., 5., 25., 25.]]), np.array([0])),
    }
    preds = {
        'a.jpg': np.array([[0, 0.9, 10, 10, 50, 50],
                           [1, 0.8, 60, 60, 90, 90]]),
        'b.jpg': np.array([[0, 0.95, 5, 5, 25, 25]]),
    }
    res = voc_map(gt, preds)
    assert res['map'] == pytest.approx(1.0)
    assert res['ap']['hat'] == pytest.approx(1.0)
    assert res['ap']['person'] == pytest.approx(1.0)


def test_false_positives_lower_ap():
    gt = {'a.jpg': (np.array([[10., 10., 50., 50.]]), np.array([0]))}
    preds = {'a.jpg': np.array([[0, 0.9, 10, 10, 50, 50],
                                [0, 0.95, 200, 200, 250, 250]])}
    res = voc_map(gt, preds)
    # highest-scored det is a FP -> precision at recall 1 is 0.5
    assert res['ap']['hat'] == pytest.approx(0.5)


def test_missed_gt_lowers_recall():
    gt = {'a.jpg': (np.array([[10., 10., 50., 50.], [100., 100., 150., 150.]]),
                    np.array([0, 0]))}
    preds = {'a.jpg': np.array([[0, 0.9, 10, 10, 50, 50]])}
    res = voc_map(gt, preds)
    assert res['ap']['hat'] == pytest.approx(0.5)


def test_duplicate_detection_is_fp():
    gt = {'a.jpg': (np.array([[10., 10., 50., 50.]]), np.array([0]))}
    preds = {'a.jpg': np.array([[0, 0.9, 10, 10, 50, 50],
                                [0, 0.8, 11, 11, 51, 51]])}
    res = voc_map(gt, preds)
    assert res['ap']['hat'] == pytest.approx(1.0)  # dup ranked below the TP


def test_average_precision_shape():
    ap = average_precision(np.array([0.5, 1.0]), np.array([1.0, 0.5]))
    assert 0.0 < ap <= 1.0


def test_nms_eager_basics():
    boxes = torch.tensor([[0., 0., 10., 10.],
                          [1., 1., 11., 11.],     # overlaps 1st
                          [50., 50., 60., 60.]])
    scores = torch.tensor([0.9, 0.8, 0.7])
    keep = eager.nms(boxes, scores, 0.5)
    assert keep.tolist() == [0, 2]
    # high threshold keeps everything
    keep2 = eager.nms(boxes, scores, 0.99)
    assert keep2.tolist() == [0, 1, 2]


def test_nms_matches_torchvision_if_available():
    try:
        import torchvision
    except Exception:
        pytest.skip('torchvision unavailable')
    torch.manual_seed(0)
    xy = torch.rand(64, 2) * 100
    wh = torch.rand(64, 2) * 30
    boxes = torch.cat([xy, xy + wh], dim=1)
    scores = torch.rand(64)
    got = eager.nms(boxes, scores, 0.5)
    want = torchvision.ops.nms(boxes, scores, 0.5)
    assert got.tolist() == want.tolist()


def test_soft_nms_decays_overlaps():
    boxes = torch.tensor([[0., 0., 10., 10.],
                          [0., 0., 10., 10.],
                          [50., 50., 60., 60.]])
    scores = torch.tensor([0.9, 0.85, 0.7])
    keep, rescored = eager.soft_nms(boxes, scores, score_th=0.01)
    assert 0 in keep.tolist() and 2 in keep.tolist()
    # identical box got decayed hard
    if 1 in keep.tolist():
        i = keep.tolist().index(1)
        assert rescored[i] < 0.4


def test_resize_box_to_original_scale():
    out = resize_box_to_original_scale(
        np.array([[10., 20., 30., 40.]]), (200, 100), (100, 100))
    np.testing.assert_allclose(out, [[20., 20., 60., 40.]])


def test_write_detection_txt(tmp_path):
    preds = {'img1.jpg': np.array([[0, 0.9, 10, 20, 30, 40]]),
             'img2.jpg': np.zeros((0, 6))}
    write_detection_txt(str(tmp_path), preds)
    txt = (tmp_path / 'img1.txt').read_text().strip()
    assert txt == '0 0.900000 10 20 30 40'
    assert (tmp_path / 'img2.txt').read_text() == ''


def test_nms_batched_eager_matches_per_image():
    """Eager batched-NMS twin == filter-then-NMS per image (the CPU
    Prediction path now routes through it)."""
    import torch
    from real_time_helmet_detection_amd.ops import eager
    torch.manual_seed(9)
    B, N = 3, 60
    ctr = torch.rand(B, N, 2) * 200
    wh2 = torch.rand(B, N, 2) * 40 + 4
    boxes = torch.cat([ctr - wh2, ctr + wh2], dim=2)
    scores = torch.rand(B, N)
    idx, counts = eager.nms_batched(boxes, scores, 0.5, 0.3)
    for i in range(B):
        keep_conf = scores[i] >= 0.3
        sel = keep_conf.nonzero(as_tuple=False).squeeze(1)
        want = sel[eager.nms(boxes[i][sel], scores[i][sel], 0.5)]
        assert idx[i, :counts[i]].tolist() == want.tolist()
