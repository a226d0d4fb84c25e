"""Coverage for the viz/normalizer utilities (reference utils.py:33-94) and
the amp module's context managers."""

import numpy as np
import pytest
import torch

from real_time_helmet_detection_amd import amp
from real_time_helmet_detection_amd.utils import (
    AverageMeter, get_normalizer, ten2pil, draw_box, write_text,
    blend_heatmap, save_pickle, load_pickle)


def test_average_meter():
    m = AverageMeter()
    for v in (1.0, 2.0, 3.0):
        m.update(v)
    assert m.avg == pytest.approx(2.0)
    assert m.sum == pytest.approx(6.0)
    m.reset()
    assert m.count == 0 and m.avg == 0


def test_normalizer_roundtrip():
    norm = get_normalizer('imagenet')
    denorm = get_normalizer('imagenet', inverse=True)
    x = torch.rand(3, 8, 8)
    back = denorm(norm(x))
    assert torch.allclose(back, x, atol=1e-5)
    # scratch mode uses 0.5/0.5
    n2 = get_normalizer('scratch')
    y = n2(torch.full((3, 2, 2), 0.5))
    assert torch.allclose(y, torch.zeros(3, 2, 2), atol=1e-6)


def test_ten2pil_and_drawing():
    norm = get_normalizer('imagenet')
    img = norm(torch.rand(3, 32, 32))
    pil = ten2pil(img, 'imagenet')
    assert pil.size == (32, 32)
    pil = draw_box(pil, [4, 4, 20, 20], cls=0)
    pil = draw_box(pil, [8, 8, 28, 28], cls=1)
    pil = write_text(pil, 'hat: 0.92', (4, 2), fontsize=8)
    assert pil.size == (32, 32)


def test_blend_heatmap():
    norm = get_normalizer('imagenet')
    img = norm(torch.rand(3, 64, 64))
    hm = torch.rand(2, 16, 16)   # num_cls heatmap at stride 4
    out = blend_heatmap(img, hm, 'imagenet')
    assert out.size == (64, 64)


def test_pickle_roundtrip(tmp_path):
    p = str(tmp_path / 'x.pickle')
    save_pickle(p, {'a': 1, 'b': [1, 2]})
    assert load_pickle(p) == {'a': 1, 'b': [1, 2]}


def test_autocast_nesting_cpu():
    assert not amp.is_autocast_enabled()
    with amp.autocast(enabled=True):
        assert amp.is_autocast_enabled()
        with amp.autocast(enabled=False):
            assert not amp.is_autocast_enabled()
        assert amp.is_autocast_enabled()
    assert not amp.is_autocast_enabled()


def test_fp8_autocast_flag():
    assert not amp.fp8_enabled()
    with amp.fp8_autocast():
        assert amp.fp8_enabled()
    assert not amp.fp8_enabled()


def test_grad_scaler_api_compat():
    """Reference train.py used torch GradScaler's scale/step/update; our
    bf16 no-op scaler keeps the surface."""
    sc = amp.GradScaler()
    w = torch.nn.Parameter(torch.ones(3))
    opt = torch.optim.SGD([w], lr=0.1)
    loss = (w * 2).sum()
    sc.scale(loss).backward()
    sc.step(opt)
    sc.update()
    assert w.grad is not None
    sd = sc.state_dict()
    sc.load_state_dict(sd)


def test_draw_box_degenerate():
    """Inverted-corner boxes (undertrained predictions) must render, not
    raise (PIL requires x1 >= x0)."""
    from PIL import Image
    from real_time_helmet_detection_amd.utils import draw_box
    img = Image.new('RGB', (64, 64))
    draw_box(img, [40.0, 40.0, 10.0, 12.0], cls=1)
    draw_box(img, [5, 5, 5, 5], cls=0)
