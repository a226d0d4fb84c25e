"""Model zoo tests: param-count parity with the reference (SURVEY.md §2.2
measured ground truth), output shapes, block variants."""

import pytest
import torch

from real_time_helmet_detection_amd.models import StackedHourglass, Residual


def n_params(m):
    return sum(p.numel() for p in m.parameters())


def test_param_count_published_config():
    # reference measured: 4.98 M params for num_stack=1, in_ch=128, out 6
    m = StackedHourglass(num_stack=1, in_ch=128, out_ch=6)
    assert n_params(m) == 4984070


def test_param_count_two_stack():
    m = StackedHourglass(num_stack=2, in_ch=128, out_ch=6)
    assert n_params(m) == 9154956


def test_forward_shape():
    m = StackedHourglass(num_stack=2, in_ch=64, out_ch=6).eval()
    with torch.no_grad():
        y = m(torch.randn(2, 3, 128, 128))
    assert y.shape == (2, 2, 6, 32, 32)


def test_backward_runs():
    m = StackedHourglass(num_stack=1, in_ch=32, out_ch=6)
    y = m(torch.randn(2, 3, 64, 64))
    y.sum().backward()
    grads = [p.grad for p in m.parameters()]
    assert all(g is not None for g in grads)


@pytest.mark.parametrize('act', ['ReLU', 'LReLU', 'PReLU', 'Linear', 'Mish',
                                 'CELU'])
def test_activation_variants(act):
    m = StackedHourglass(num_stack=1, in_ch=16, out_ch=6, activation=act).eval()
    with torch.no_grad():
        y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 1, 6, 16, 16)


@pytest.mark.parametrize('pool', ['Max', 'Avg', 'Conv'])
def test_pool_variants(pool):
    m = StackedHourglass(num_stack=1, in_ch=16, out_ch=6, pool=pool).eval()
    with torch.no_grad():
        y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 1, 6, 16, 16)


def test_neck_spp():
    m = StackedHourglass(num_stack=1, in_ch=16, out_ch=6, neck_pool='SPP').eval()
    with torch.no_grad():
        y = m(torch.randn(1, 3, 64, 64))
    assert y.shape == (1, 1, 6, 16, 16)


def test_increase_ch():
    m = StackedHourglass(num_stack=2, in_ch=128, out_ch=6, increase_ch=128)
    # 84.95 M params measured on the reference
    assert n_params(m) == pytest.approx(84.95e6, rel=0.01)


def test_residual_skip_projection():
    r = Residual(8, 16)
    assert not isinstance(r.skip, torch.nn.Identity)
    r2 = Residual(16, 16)
    assert isinstance(r2.skip, torch.nn.Identity)


def test_trace_equivalence():
    # reference hourglass.py:250-256 self-test: jit trace == eager
    m = StackedHourglass(num_stack=1, in_ch=16, out_ch=6).eval()
    x = torch.randn(1, 3, 64, 64)
    with torch.no_grad():
        traced = torch.jit.trace(m, x)
        x2 = torch.ones(1, 3, 64, 64)
        torch.testing.assert_close(m(x2), traced(x2))


def test_state_dict_key_layout():
    # attribute naming keeps reference checkpoints loadable
    m = StackedHourglass(num_stack=1, in_ch=16, out_ch=6)
    keys = set(m.state_dict().keys())
    assert 'pre_layer.layers.0.convolution.weight' in keys
    assert 'hourglass_lst.0.up1.conv1.convolution.weight' in keys
    assert 'neck_lst.0.layers.1.bn.running_mean' in keys
    assert 'head_lst.0.layer.convolution.bias' in keys


def test_relu_actually_applied():
    """Regression: the eager path once dropped the fused activation string —
    a ReLU model's block outputs must be non-negative."""
    from real_time_helmet_detection_amd.models import Convolution, Residual
    torch.manual_seed(0)
    m = Convolution(8, 8, 3, bias=False, bn=True, activation='ReLU').eval()
    x = torch.randn(2, 8, 16, 16)
    with torch.no_grad():
        y = m(x)
    assert (y >= 0).all()
    assert (y > 0).any()
    r = Residual(8, 8).eval()
    with torch.no_grad():
        yr = r(x)
    assert (yr >= 0).all()


def test_lrelu_slope():
    from real_time_helmet_detection_amd.models import Convolution
    import torch.nn.functional as F
    torch.manual_seed(1)
    m = Convolution(8, 8, 1, bias=True, bn=False, activation='LReLU').eval()
    x = torch.randn(2, 8, 4, 4)
    with torch.no_grad():
        y = m(x)
        want = F.leaky_relu(
            F.conv2d(x, m.convolution.weight, m.convolution.bias), 0.01)
    torch.testing.assert_close(y, want)
