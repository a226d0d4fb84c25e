"""GPU coverage for the reference's non-default configurations: multi-stack
deep supervision, SPP pooling, Mish/PReLU activations, normalized
coordinates, increase_ch, soft-NMS (reference config.py flags; hourglass.py
blocks). Each case checks CPU-eager vs GPU-HIP parity or basic sanity."""

import copy

import pytest
import torch

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def _pair(seed=0, **kw):
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(seed)
    cpu = StackedHourglass(**kw)
    gpu = copy.deepcopy(cpu).cuda().to(memory_format=CL)
    return cpu, gpu


def rel(got, want):
    got, want = got.detach().float().cpu(), want.detach().float().cpu()
    return ((got - want).abs().max() / want.abs().max().clamp(min=1e-6))


def _check(cpu, gpu, size=128, tol=3e-3):
    x = torch.randn(2, 3, size, size)
    cpu.eval(), gpu.eval()
    with torch.no_grad():
        yc = cpu(x)
        yg = gpu(x.cuda().contiguous(memory_format=CL))
    assert yc.shape == yg.shape
    assert rel(yg, yc) < tol, rel(yg, yc)


def test_two_stack_parity():
    cpu, gpu = _pair(num_stack=2, in_ch=32, out_ch=6)
    _check(cpu, gpu)


def test_increase_ch_parity():
    cpu, gpu = _pair(num_stack=1, in_ch=32, out_ch=6, increase_ch=32)
    _check(cpu, gpu)


def test_spp_neck_parity():
    cpu, gpu = _pair(num_stack=1, in_ch=32, out_ch=6, neck_pool='SPP')
    _check(cpu, gpu)


def test_mish_activation_parity():
    cpu, gpu = _pair(num_stack=1, in_ch=32, out_ch=6, activation='Mish')
    _check(cpu, gpu)


def test_prelu_activation_parity():
    cpu, gpu = _pair(num_stack=1, in_ch=32, out_ch=6, activation='PReLU')
    _check(cpu, gpu)


def test_avg_pool_parity():
    cpu, gpu = _pair(num_stack=1, in_ch=32, out_ch=6, pool='Avg')
    _check(cpu, gpu)


def test_two_stack_backward_parity():
    cpu, gpu = _pair(num_stack=2, in_ch=32, out_ch=6, seed=3)
    x = torch.randn(2, 3, 64, 64)
    xg = x.cuda().contiguous(memory_format=CL)
    cpu.train(), gpu.train()
    cpu(x).float().pow(2).mean().backward()
    gpu(xg).float().pow(2).mean().backward()
    # fp32 noise amplifies through ~50 BN layers of the 2-stack chain and
    # conv-bias-under-BN grads are analytically zero, so per-parameter
    # relative metrics are meaningless at depth (see
    # test_model_backward_parity_fp32's floor) — assert on the GLOBAL
    # gradient vector instead, which still catches any structural bug
    # (a missing grad path would contribute its full norm).
    name_cgrad = {n: p.grad for n, p in cpu.named_parameters()}
    cs, gs = [], []
    for n, p in gpu.named_parameters():
        if p.grad is None or name_cgrad[n] is None:
            continue
        cs.append(name_cgrad[n].float().flatten())
        gs.append(p.grad.float().cpu().flatten())
    c = torch.cat(cs)
    g = torch.cat(gs)
    rel_l2 = (g - c).norm() / c.norm()
    # CPU-vs-GPU reduction orders differ and the noise amplifies through
    # the ~50-BN-layer double-depth chain (2.7e-2..6.5e-2 measured). A
    # structural bug (missing grad path) would contribute its full norm,
    # i.e. O(1) — assert well below that; the companion convergence test
    # covers functional correctness.
    assert rel_l2 < 0.15, rel_l2.item()


def test_two_stack_training_converges():
    """Functional check: a 2-stack model trains on GPU (loss decreases),
    covering the merge_feature/merge_prediction path end to end."""
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd import amp
    torch.manual_seed(9)
    net = StackedHourglass(2, 32, 6).cuda().to(memory_format=CL).train()
    opt = torch.optim.Adam(net.parameters(), lr=1e-3)
    x = torch.randn(2, 3, 64, 64, device='cuda').contiguous(
        memory_format=CL)
    tgt = torch.randn(2, 2, 6, 16, 16, device='cuda')
    losses = []
    for _ in range(30):
        opt.zero_grad(set_to_none=True)
        with amp.autocast(enabled=True):
            y = net(x)
        loss = (y.float() - tgt).pow(2).mean()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < 0.7 * losses[0], (losses[0], losses[-1])


def test_normalized_coord_decode():
    from real_time_helmet_detection_amd.transform import hm2box
    torch.manual_seed(5)
    hm = torch.sigmoid(torch.randn(2, 32, 32, device='cuda'))
    off = torch.sigmoid(torch.randn(2, 32, 32, device='cuda'))
    wh = torch.sigmoid(torch.randn(2, 32, 32, device='cuda'))
    b, c, s = hm2box(hm, off, wh, scale_factor=4, topk=20, conf_th=0.0,
                     normalized=True)
    bc, cc, sc = hm2box(hm.cpu(), off.cpu(), wh.cpu(), scale_factor=4,
                        topk=20, conf_th=0.0, normalized=True)
    assert torch.allclose(s.cpu(), sc, atol=1e-5)
    assert torch.allclose(b.cpu(), bc, atol=1e-3)


def test_soft_nms_prediction_gpu():
    from real_time_helmet_detection_amd.engine.evaluator import Prediction
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(6)
    net = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL).eval()
    pred = Prediction(net, topk=30, scale_factor=4, conf_th=0.05,
                      nms='soft-nms', nms_th=0.5).cuda()
    x = torch.randn(1, 3, 128, 128, device='cuda').contiguous(
        memory_format=CL)
    with torch.no_grad():
        boxes, clss, scores = pred(x)
    assert len(boxes) == 1 and torch.isfinite(boxes[0]).all()
