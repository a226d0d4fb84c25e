"""GPU end-to-end tests: model parity vs CPU eager, training step, decode
pipeline, amp. Requires the in-tree gfx950 extension."""

import copy

import pytest
import torch

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def _models(in_ch=32, num_stack=1, seed=0):
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(seed)
    cpu = StackedHourglass(num_stack=num_stack, in_ch=in_ch, out_ch=6)
    gpu = copy.deepcopy(cpu).cuda().to(memory_format=CL)
    return cpu, gpu


def rel_err(got, want):
    got = got.detach().float().cpu()
    want = want.detach().float().cpu()
    return ((got - want).abs().max() /
            want.abs().max().clamp(min=1e-6)).item()


def test_model_forward_parity_eval_fp32():
    cpu, gpu = _models()
    cpu.eval()
    gpu.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        want = cpu(x)
        got = gpu(x.cuda().contiguous(memory_format=CL))
    assert rel_err(got, want) < 2e-4


def test_model_forward_parity_train_fp32():
    cpu, gpu = _models(seed=1)
    cpu.train()
    gpu.train()
    x = torch.randn(4, 3, 64, 64)
    with torch.no_grad():
        want = cpu(x)
        got = gpu(x.cuda().contiguous(memory_format=CL))
    assert rel_err(got, want) < 5e-4
    # BN running stats tracked identically
    for (k1, b1), (k2, b2) in zip(cpu.named_buffers(), gpu.named_buffers()):
        assert k1 == k2
        assert rel_err(b2, b1) < 5e-4, k1


def test_model_backward_parity_fp32():
    cpu, gpu = _models(seed=2)
    cpu.train()
    gpu.train()
    x = torch.randn(4, 3, 64, 64)
    cpu(x).float().pow(2).mean().backward()
    gpu(x.cuda().contiguous(memory_format=CL)).float().pow(2).mean().backward()
    # relative-L2 metric: tiny forward fp differences amplify through ~25
    # BN layers; per-element max-rel is meaningless for near-zero grads
    # (e.g. conv bias under BN, analytically 0)
    worst = {}
    ref_scale = max(pc.grad.abs().max().item()
                    for pc in cpu.parameters() if pc.grad is not None)
    for (k, pc), (_, pg) in zip(cpu.named_parameters(),
                                gpu.named_parameters()):
        if pc.grad is None:
            assert pg.grad is None, k
            continue
        diff = (pg.grad.cpu().float() - pc.grad.float())
        denom = pc.grad.float().norm().clamp(min=1e-4 * ref_scale)
        worst[k] = (diff.norm() / denom).item()
    bad = {k: v for k, v in worst.items() if v > 5e-2}
    assert not bad, f'grad mismatches (rel-L2): {bad}'


def test_model_forward_bf16_amp_close():
    from real_time_helmet_detection_amd import amp
    cpu, gpu = _models(seed=3)
    cpu.eval()
    gpu.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        want = cpu(x)
        with amp.autocast(True):
            got = gpu(x.cuda().contiguous(memory_format=CL))
    assert rel_err(got, want) < 0.1  # bf16 through 20+ layers


def test_train_step_loss_decreases():
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.loss import LossCalculator
    from real_time_helmet_detection_amd.engine.trainer import \
        compute_stack_losses
    from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
    from real_time_helmet_detection_amd import amp

    torch.manual_seed(0)
    ds = SyntheticVOC(transform=TestAugmentor(128), size=2, imsize=128,
                      seed=3)
    img, hm, off, wh, mask, _ = ds.collate_fn([ds[0], ds[1]])
    img = img.cuda().contiguous(memory_format=CL)
    hm, off, wh, mask = (t.cuda() for t in (hm, off, wh, mask))

    net = StackedHourglass(1, 128, 6).cuda().to(memory_format=CL)
    calc = LossCalculator().cuda()
    opt = torch.optim.Adam(net.parameters(), lr=1e-3)
    losses = []
    for i in range(20):
        opt.zero_grad(set_to_none=True)
        with amp.autocast(True):
            out = net(img)
        total, _ = compute_stack_losses(out, calc, hm, off, wh, mask, 2,
                                        False)
        total.backward()
        opt.step()
        losses.append(total.item())
    assert all(torch.isfinite(torch.tensor(losses)).tolist())
    assert losses[-1] < losses[0]


def test_prediction_pipeline_gpu():
    from real_time_helmet_detection_amd.engine.evaluator import Prediction
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(4)
    net = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL).eval()
    pred = Prediction(net, topk=50, scale_factor=4, conf_th=0.1, nms='nms',
                      nms_th=0.5).cuda()
    x = torch.randn(2, 3, 128, 128).cuda().contiguous(memory_format=CL)
    with torch.no_grad():
        boxes, clss, scores = pred(x)
    assert len(boxes) == 2
    for b, c, s in zip(boxes, clss, scores):
        assert b.shape[1] == 4 if b.numel() else True
        assert torch.isfinite(b).all()


def test_native_extension_is_loaded():
    """Guard against silent eager fallback: the .so must be loaded and the
    conv path must be our kernel (monkeypatch-detect by op error type)."""
    from real_time_helmet_detection_amd.ops import _backend
    ext = _backend.require_ext()
    assert ext.__file__.endswith('.so')
    import real_time_helmet_detection_amd.ops as ops
    assert ops.available()


def test_sub_division_accumulation_gpu():
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(5)
    netA = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL)
    netB = copy.deepcopy(netA)
    x = torch.randn(4, 3, 64, 64).cuda().contiguous(memory_format=CL)
    netA(x).float().mean().backward()
    (netB(x[:2]).float().mean() / 2).backward()
    (netB(x[2:]).float().mean() / 2).backward()
    for pa, pb in zip(netA.parameters(), netB.parameters()):
        # BN batch stats differ between full batch and micro-batches, so
        # tolerances are loose; this checks accumulation plumbing, not BN.
        assert torch.isfinite(pb.grad).all()


def test_conv_pool_backward_gpu():
    """pool='Conv' (k2 s2) backward: stride-2 dgrad via tap decomposition."""
    from real_time_helmet_detection_amd.models import Pool
    import torch.nn.functional as F
    torch.manual_seed(6)
    p_cpu = Pool(32, 'Conv')
    p_gpu = copy.deepcopy(p_cpu).cuda().to(memory_format=CL)
    x = torch.randn(2, 32, 16, 16)
    xg = x.cuda().contiguous(memory_format=CL).requires_grad_(True)
    xc = x.clone().requires_grad_(True)
    y_cpu = p_cpu(xc)
    y_gpu = p_gpu(xg)
    dy = torch.randn_like(y_cpu)
    y_cpu.backward(dy)
    y_gpu.backward(dy.cuda().contiguous(memory_format=CL))
    assert rel_err(y_gpu, y_cpu) < 1e-4
    assert rel_err(xg.grad, xc.grad) < 1e-4
    assert rel_err(p_gpu.pool.weight.grad, p_cpu.pool.weight.grad) < 1e-3


def test_fp8_inference_close_to_bf16():
    # in_ch=128 so the convs actually take the fp8 K=128 path (cin%128)
    from real_time_helmet_detection_amd import amp
    torch.manual_seed(7)
    _, gpu = _models(in_ch=128, seed=7)
    gpu.eval()
    x = torch.randn(2, 3, 64, 64).cuda().contiguous(memory_format=CL)
    with torch.no_grad():
        with amp.autocast(True):
            ref = gpu(x)
        with amp.autocast(True), amp.fp8_autocast(True):
            got = gpu(x)
    # e4m3 weights+activations through ~25 layers: loose tolerance
    assert rel_err(got, ref) < 0.25
    assert torch.isfinite(got).all()


def test_graphed_train_step():
    """GraphedTrainStep: replays must actually train (loss decreases), the
    per-iteration loss log must hold REAL values (a capture records
    without executing — the first logged entry must come from a replay,
    not the uninitialized static buffers), and shapes beyond the capture
    key must fall back cleanly."""
    from real_time_helmet_detection_amd.engine.graphed import \
        GraphedTrainStep
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.loss import LossCalculator
    from real_time_helmet_detection_amd.data import (SyntheticVOC,
                                                     TestAugmentor)
    torch.manual_seed(31)
    net = StackedHourglass(1, 64, 6).cuda().to(memory_format=CL)
    calc = LossCalculator().cuda()
    opt = torch.optim.Adam(net.parameters(), lr=2e-3, fused=True,
                           capturable=True)
    ds = SyntheticVOC(transform=TestAugmentor(128), size=4, imsize=128,
                      seed=5)
    img, hm, off, wh, mask, _ = ds.collate_fn([ds[i] for i in range(4)])
    batch = tuple(t.cuda() for t in (img.contiguous(memory_format=CL),
                                     hm, off, wh, mask))

    g = GraphedTrainStep(net, calc, opt, num_cls=2, normalized_coord=False,
                         amp_on=True)
    net.train()
    for _ in range(30):
        out = g.step(*batch)
        assert out is not None, 'graph capture failed'
    assert g.enabled and len(g.graphs) == 1
    calc.flush_log()
    totals = calc.log['total']
    # 30 replays logged (warmup bypasses the log), all real finite values
    assert len(totals) == 30
    assert all(abs(v) < 1e4 for v in totals), totals[:5]
    assert totals[-1] < totals[0] * 0.9, (totals[0], totals[-1])

    # an unseen shape past MAX_SHAPES returns None (eager fallback)
    g.MAX_SHAPES = 1
    small = tuple(t[:, :, :64, :64].contiguous() if t.dim() == 4 else t
                  for t in batch)
    assert g.step(*small) is None


def test_graphed_predictor_matches_eager():
    from real_time_helmet_detection_amd.engine.evaluator import (
        Prediction, GraphedPredictor)
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(8)
    net = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL).eval()
    pred = Prediction(net, topk=20, scale_factor=4, conf_th=0.1, nms='nms',
                      nms_th=0.5).cuda()
    x = torch.randn(2, 3, 128, 128).cuda().contiguous(memory_format=CL)
    g = GraphedPredictor(pred, x.clone())
    from real_time_helmet_detection_amd import amp
    with torch.no_grad(), amp.autocast(True):
        want = pred(x)
    got = g(x)
    for wb, gb in zip(want[0], got[0]):
        assert wb.shape == gb.shape
        torch.testing.assert_close(gb, wb, rtol=0.05, atol=0.5)


def test_overfit_synthetic_map():
    """Training correctness end-to-end: overfit 4 synthetic images until the
    detector finds their boxes (mAP through the FULL pipeline: HIP train ->
    decode -> NMS -> VOC mAP)."""
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.loss import LossCalculator
    from real_time_helmet_detection_amd.engine.trainer import \
        compute_stack_losses
    from real_time_helmet_detection_amd.engine.evaluator import Prediction
    from real_time_helmet_detection_amd.engine.metrics import voc_map
    from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
    from real_time_helmet_detection_amd import amp
    import numpy as np

    torch.manual_seed(0)
    ds = SyntheticVOC(transform=TestAugmentor(256), size=4, imsize=256,
                      seed=11)
    items = [ds[i] for i in range(4)]
    img, hm, off, wh, mask, dicts = ds.collate_fn(items)
    img = img.cuda().contiguous(memory_format=CL)
    hm, off, wh, mask = (t.cuda() for t in (hm, off, wh, mask))

    net = StackedHourglass(1, 64, 6).cuda().to(memory_format=CL)
    calc = LossCalculator().cuda()
    opt = torch.optim.Adam(net.parameters(), lr=2e-3)
    net.train()
    for i in range(500):
        opt.zero_grad(set_to_none=True)
        with amp.autocast(True):
            out = net(img)
        total, _ = compute_stack_losses(out, calc, hm, off, wh, mask, 2,
                                        False)
        total.backward()
        opt.step()

    net.eval()
    pred = Prediction(net, topk=20, scale_factor=4, conf_th=0.15, nms='nms',
                      nms_th=0.5).cuda()
    with torch.no_grad():
        boxes, clss, scores = pred(img)

    gt, preds = {}, {}
    for i in range(4):
        _, gtb, gtl, voc = items[i]
        name = voc['annotation']['filename']
        gt[name] = (np.asarray(gtb, np.float64), np.asarray(gtl))
        b = boxes[i].cpu().numpy()
        c = clss[i].cpu().numpy()[:, None]
        s = scores[i].cpu().numpy()[:, None]
        preds[name] = (np.hstack([c, s, b]) if len(b)
                       else np.zeros((0, 6)))
    res = voc_map(gt, preds)
    assert res is not None
    # overfit on 4 images: the detector must find most boxes
    assert res['map'] > 0.35, f"mAP after overfit: {res}"


def test_native_traced_export_parity(tmp_path):
    """export_model's GPU trace records torch.ops.rthd.* (native gfx950
    kernels); the saved module must reload and reproduce the live
    predictor's detections (reference export.py:132-152 do_test)."""
    from real_time_helmet_detection_amd.engine.exporter import (
        Export, export_model)
    from real_time_helmet_detection_amd.models import StackedHourglass
    torch.manual_seed(7)
    net = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL).eval()
    pred = Export(net, topk=50, scale_factor=4, conf_th=0.05,
                  nms_th=0.5).cuda()
    paths = export_model(pred, save_dir=str(tmp_path), imsize=128,
                         do_gpu=True, native=True)
    assert 'gpu' in paths
    loaded = torch.jit.load(paths['gpu'])
    g = str(loaded.inlined_graph)
    assert 'rthd::conv_fwd' in g, 'native ops not in traced graph'

    # export_model moved pred through .cpu() and back; rebuild a GPU input
    pred = pred.cuda()
    x = torch.randn(1, 3, 128, 128, device='cuda')
    with torch.no_grad():
        want = pred(x)
        got = loaded(x)
    for w, g_ in zip(want, got):
        assert w.shape == g_.shape
        if w.numel():
            assert torch.allclose(w.float().cpu(), g_.float().cpu(),
                                  atol=2e-2, rtol=2e-2)


def test_infer_cache_invalidation():
    """The packed-weight/scale-shift inference cache (keyed by parameter
    versions) must refresh when weights or BN stats change."""
    from real_time_helmet_detection_amd.models.hourglass import Convolution
    torch.manual_seed(11)
    mod = Convolution(16, 16, 3, bn=True, activation='ReLU').cuda() \
        .to(memory_format=CL).eval()
    x = torch.randn(2, 16, 16, 16, device='cuda').contiguous(
        memory_format=CL)
    with torch.no_grad():
        y1 = mod(x)
        mod.convolution.weight.mul_(2.0)   # bump version in-place
        y2 = mod(x)
        mod.bn.running_var.mul_(4.0)
        y3 = mod(x)
    assert not torch.allclose(y1, y2)
    assert not torch.allclose(y2, y3)
    # fresh module with identical params reproduces y3
    import copy
    mod2 = copy.deepcopy(mod)
    if hasattr(mod2.convolution, '_rthd_infer_cache'):
        del mod2.convolution._rthd_infer_cache
    with torch.no_grad():
        y4 = mod2(x)
    assert torch.allclose(y3, y4)


def test_full_backward_deterministic():
    """Whole train-step gradient chain is bitwise reproducible: wgrad
    partial writeback, BN staged reductions, and the loss partial sums are
    all fixed-order (no same-address float atomics)."""
    from real_time_helmet_detection_amd.models import StackedHourglass
    from real_time_helmet_detection_amd.ops import functional as F2
    from real_time_helmet_detection_amd import amp
    torch.manual_seed(21)
    net = StackedHourglass(1, 32, 6).cuda().to(memory_format=CL).train()
    x = torch.randn(2, 3, 64, 64, device='cuda').contiguous(
        memory_format=CL)

    def grads():
        # fresh BN running stats each pass so the two passes are identical
        net2 = copy.deepcopy(net)
        for p in net2.parameters():
            p.grad = None
        with amp.autocast(enabled=True):
            y = net2(x)
        y.float().pow(2).mean().backward()
        return [p.grad.clone() for p in net2.parameters()
                if p.grad is not None]

    g1 = grads()
    g2 = grads()
    assert len(g1) == len(g2)
    for a, b in zip(g1, g2):
        assert torch.equal(a, b)
