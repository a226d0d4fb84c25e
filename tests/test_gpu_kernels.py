"""HIP kernel parity tests vs fp32 eager oracles (run on MI355X).

Every custom gfx950 kernel is compared against the plain PyTorch fp32
reference op computed on CPU (SURVEY.md §4 test strategy). bf16 paths use
relative-norm tolerances sized for bf16 accumulation; fp32 MFMA paths are
exact-f32 (fmaf-chain numerics) and use tight tolerances.
"""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def _C():
    from real_time_helmet_detection_amd.ops import _backend
    return _backend.require_ext()


def to_gpu(x, dtype=torch.float32):
    return x.to('cuda', dtype).contiguous(memory_format=CL)


def rel_err(got, want):
    got = got.detach().float().cpu()
    want = want.detach().float().cpu()
    denom = want.abs().max().clamp(min=1e-6)
    return ((got - want).abs().max() / denom).item()


# ---------------------------------------------------------------- add_act --

@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('act', [0, 1, 2])
def test_add_act_fwd_bwd(dtype, act):
    torch.manual_seed(0)
    a = torch.randn(2, 32, 16, 16)
    b = torch.randn(2, 32, 16, 16)
    if dtype == torch.bfloat16:
        # reference on bf16-rounded values so act masks match at boundaries
        a = a.to(dtype).float()
        b = b.to(dtype).float()
    z = a + b
    want = z if act == 0 else (F.relu(z) if act == 1
                               else F.leaky_relu(z, 0.01))
    got = _C().add_act_fwd(to_gpu(a, dtype), to_gpu(b, dtype), act)
    tol = 0.02 if dtype == torch.bfloat16 else 1e-6
    assert rel_err(got, want) < tol

    dy = torch.randn_like(z)
    grad = torch.ones_like(z)
    if act == 1:
        grad = (want > 0).float()
    elif act == 2:
        grad = torch.where(want > 0, 1.0, 0.01)
    want_dz = dy * grad
    got_dz = _C().add_act_bwd(to_gpu(dy, dtype), got, act)
    assert rel_err(got_dz, want_dz) < (0.02 if dtype == torch.bfloat16
                                       else 1e-6)


# ------------------------------------------------------------------ pools --

@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_maxpool2x2(dtype):
    torch.manual_seed(1)
    x = torch.randn(2, 32, 16, 16)
    if dtype == torch.bfloat16:
        x = x.to(dtype).float()  # argmax ties must match the bf16 values
    want = F.max_pool2d(x, 2, 2)
    out = _C().pool2x2_fwd(to_gpu(x, dtype), True, True)
    assert rel_err(out[0], want) < (0.01 if dtype == torch.bfloat16 else 1e-6)

    xg = x.clone().requires_grad_(True)
    dy = torch.randn(2, 32, 8, 8)
    F.max_pool2d(xg, 2, 2).backward(dy)
    got_dx = _C().pool2x2_bwd(to_gpu(dy, dtype), out[1], True, 16, 16)
    assert rel_err(got_dx, xg.grad) < (0.01 if dtype == torch.bfloat16
                                       else 1e-6)


def test_avgpool2x2():
    torch.manual_seed(2)
    x = torch.randn(2, 32, 16, 16)
    want = F.avg_pool2d(x, 2, 2)
    out = _C().pool2x2_fwd(to_gpu(x), False, False)
    assert rel_err(out[0], want) < 1e-6


@pytest.mark.parametrize('k', [3, 5, 9, 13])
def test_maxpool_same(k):
    torch.manual_seed(3)
    x = torch.randn(2, 32, 16, 16)
    want = F.max_pool2d(x, k, stride=1, padding=k // 2)
    out = _C().maxpool_same_fwd(to_gpu(x), k, True)
    assert rel_err(out[0], want) < 1e-6

    xg = x.clone().requires_grad_(True)
    dy = torch.randn_like(x)
    F.max_pool2d(xg, k, stride=1, padding=k // 2).backward(dy)
    got_dx = _C().maxpool_same_bwd(to_gpu(dy), out[1], k)
    assert rel_err(got_dx, xg.grad) < 1e-5


# --------------------------------------------------------------- upsample --

def test_upsample2x_add():
    torch.manual_seed(4)
    x = torch.randn(2, 32, 8, 8)
    skip = torch.randn(2, 32, 16, 16)
    want = F.interpolate(x, scale_factor=2, mode='nearest') + skip
    got = _C().upsample2x_add_fwd(to_gpu(x), to_gpu(skip))
    assert rel_err(got, want) < 1e-6

    dy = torch.randn(2, 32, 16, 16)
    xg = x.clone().requires_grad_(True)
    F.interpolate(xg, scale_factor=2, mode='nearest').backward(dy)
    got_dx = _C().upsample2x_bwd(to_gpu(dy))
    assert rel_err(got_dx, xg.grad) < 1e-6


# ------------------------------------------------------------------- loss --

def _loss_inputs(seed=5):
    g = torch.Generator().manual_seed(seed)
    b, c, h, w = 2, 2, 32, 32
    phm = torch.rand(b, c, h, w, generator=g).clamp(1e-4, 1 - 1e-4)
    ghm = torch.rand(b, c, h, w, generator=g)
    mask = (torch.rand(b, 1, h, w, generator=g) > 0.95).float()
    poff = torch.randn(b, 2, h, w, generator=g)
    goff = torch.rand(b, 2, h, w, generator=g)
    psize = torch.randn(b, 2, h, w, generator=g) * 5
    gsize = torch.rand(b, 2, h, w, generator=g) * 10
    return phm, poff, psize, ghm, goff, gsize, mask


def test_centernet_loss_fwd_bwd():
    from real_time_helmet_detection_amd.ops import eager
    ins = _loss_inputs()
    phm, poff, psize, ghm, goff, gsize, mask = ins
    want = eager.centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                                  2.0, 4.0)
    gins = [t.cuda() for t in ins]
    losses, sums = _C().centernet_loss_fwd(*gins, 2.0, 4.0)
    for i in range(3):
        assert abs(losses[i].item() - want[i].item()) < 1e-4 * max(
            1.0, abs(want[i].item()))

    # backward vs autograd of the eager op
    phm_g = phm.clone().requires_grad_(True)
    poff_g = poff.clone().requires_grad_(True)
    psize_g = psize.clone().requires_grad_(True)
    hm_l, off_l, size_l = eager.centernet_losses(
        phm_g, poff_g, psize_g, ghm, goff, gsize, mask, 2.0, 4.0)
    (hm_l * 1.0 + off_l * 2.0 + size_l * 0.5).backward()
    gout = torch.tensor([1.0, 2.0, 0.5], device='cuda')
    dphm, dpoff, dpsize = _C().centernet_loss_bwd(
        *gins, sums, gout, 2.0, 4.0)
    assert rel_err(dphm, phm_g.grad) < 1e-3
    assert rel_err(dpoff, poff_g.grad) < 1e-5
    assert rel_err(dpsize, psize_g.grad) < 1e-5


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('sig_os', [False, True])
def test_centernet_loss_fused_logits(dtype, sig_os):
    """All-stacks fused logits kernel vs the eager per-stack sigmoid path."""
    torch.manual_seed(11)
    B, S, C, h, w = 3, 2, 2, 32, 32
    out = torch.randn(B, S, C + 4, h, w) * 2
    if dtype == torch.bfloat16:
        out = out.to(dtype).float()  # oracle on the rounded values
    ghm = torch.rand(B, C, h, w)
    goff = torch.rand(B, 2, h, w)
    gsize = torch.rand(B, 2, h, w) * 10
    mask = (torch.rand(B, 1, h, w) > 0.95).float()

    # eager oracle: per-stack sigmoid outside, fp32
    from real_time_helmet_detection_amd.ops import eager
    want = []
    out_g = out.clone().requires_grad_(True)
    total = 0
    wvec = (1.0, 2.0, 0.5)
    for s in range(S):
        o = out_g[:, s]
        phm = torch.sigmoid(o[:, :C])
        poff, psize = o[:, C:C + 2], o[:, C + 2:]
        if sig_os:
            poff, psize = torch.sigmoid(poff), torch.sigmoid(psize)
        hm_l, off_l, sz_l = eager.centernet_losses(
            phm, poff, psize, ghm, goff, gsize, mask, 2.0, 4.0)
        want.append((hm_l, off_l, sz_l))
        total = total + wvec[0] * hm_l + wvec[1] * off_l + wvec[2] * sz_l
    total.backward()

    from real_time_helmet_detection_amd.ops import hip
    out_d = out.to('cuda', dtype).requires_grad_(True)
    gts = [t.cuda() for t in (ghm, goff, gsize, mask)]
    losses = hip.centernet_losses_logits(out_d, *gts, 2.0, 4.0, sig_os)
    assert losses.shape == (S, 3)
    tol = 2e-3 if dtype == torch.bfloat16 else 1e-4
    for s in range(S):
        for i in range(3):
            w_ = want[s][i].item()
            assert abs(losses[s, i].item() - w_) < tol * max(1.0, abs(w_)), \
                (s, i, losses[s, i].item(), w_)

    wt = torch.tensor(wvec, device='cuda')
    (losses * wt).sum().backward()
    gtol = 0.03 if dtype == torch.bfloat16 else 1e-3
    assert rel_err(out_d.grad, out_g.grad) < gtol


def test_compute_stack_losses_gpu_matches_cpu():
    """The trainer's fused GPU loss path vs its eager CPU path."""
    from real_time_helmet_detection_amd.engine.trainer import \
        compute_stack_losses
    from real_time_helmet_detection_amd.loss import LossCalculator
    torch.manual_seed(12)
    B, S, C, h, w = 2, 2, 2, 32, 32
    out = torch.randn(B, S, C + 4, h, w)
    ghm = torch.rand(B, C, h, w)
    goff = torch.rand(B, 2, h, w)
    gsize = torch.rand(B, 2, h, w) * 10
    mask = (torch.rand(B, 1, h, w) > 0.95).float()

    calc_cpu = LossCalculator()
    want, _ = compute_stack_losses(out, calc_cpu, ghm, goff, gsize, mask,
                                   C, False)
    calc_gpu = LossCalculator().cuda()
    got, hm_logits = compute_stack_losses(
        out.cuda(), calc_gpu, ghm.cuda(), goff.cuda(), gsize.cuda(),
        mask.cuda(), C, False)
    assert abs(got.item() - want.item()) < 1e-3 * max(1.0, abs(want.item()))
    # returned heatmap is the last stack's LOGITS
    torch.testing.assert_close(hm_logits.cpu(), out[:, -1, :C])
    # the per-stack log entries match too
    calc_cpu.flush_log()
    calc_gpu.flush_log()
    for k in calc_cpu.LOG_KEYS:
        assert len(calc_gpu.log[k]) == len(calc_cpu.log[k]) == S
        for a, b in zip(calc_gpu.log[k], calc_cpu.log[k]):
            assert abs(a - b) < 1e-3 * max(1.0, abs(b))


# ----------------------------------------------------------------- decode --

def test_nms_overflow_falls_back_to_eager():
    """>2048 boxes exceed the LDS-resident kernel: the wrapper must fall
    back to the eager suppression instead of aborting (reference used
    torchvision.ops.nms with no cap — e.g. --topk 1000, num_stack 3)."""
    from real_time_helmet_detection_amd.ops import eager, hip
    torch.manual_seed(30)
    n = 3000
    ctr = torch.rand(n, 2) * 400
    wh2 = torch.rand(n, 2) * 60 + 5
    boxes = torch.cat([ctr - wh2, ctr + wh2], dim=1)
    scores = torch.rand(n)
    want = eager.nms(boxes, scores, 0.5)
    got = hip.nms(boxes.cuda(), scores.cuda(), 0.5)
    assert got.cpu().tolist() == want.tolist()


def test_decode_empty_heatmap():
    """No positive scores anywhere: decode must emit topk zero-score rows
    (idx 0) without tripping the peak/threshold machinery."""
    hm = torch.zeros(2, 2, 32, 32, device='cuda')
    off = torch.rand(2, 2, 32, 32, device='cuda')
    wh = torch.rand(2, 2, 32, 32, device='cuda')
    boxes, clss, scores = _C().decode_fwd(hm, off, wh, 4, 10, 3, False)
    assert scores.abs().max().item() == 0.0
    assert boxes.shape == (2, 10, 4) and torch.isfinite(boxes).all()


def test_nms_batched_global_mask_path():
    """N > 1024 exceeds the LDS mask budget — the global-scratch branch
    must match the eager result exactly."""
    from real_time_helmet_detection_amd.ops import eager
    torch.manual_seed(35)
    B, N = 2, 1500
    ctr = torch.rand(B, N, 2) * 600
    wh2 = torch.rand(B, N, 2) * 50 + 5
    boxes = torch.cat([ctr - wh2, ctr + wh2], dim=2)
    scores = torch.rand(B, N)
    want_idx, want_cnt = eager.nms_batched(boxes, scores, 0.5, 0.2)
    got_idx, got_cnt = _C().nms_batched(boxes.cuda(), scores.cuda(),
                                        0.5, 0.2)
    assert got_cnt.cpu().tolist() == want_cnt.tolist()
    for i in range(B):
        k = int(want_cnt[i])
        assert got_idx[i, :k].cpu().tolist() == want_idx[i, :k].tolist()


def test_nms_batched_matches_eager():
    """Batched NMS kernel (conf filter folded in) vs the eager per-image
    filter-then-NMS loop — exact index/count match."""
    from real_time_helmet_detection_amd.ops import eager
    torch.manual_seed(33)
    B, N = 5, 300
    ctr = torch.rand(B, N, 2) * 400
    wh2 = torch.rand(B, N, 2) * 60 + 5
    boxes = torch.cat([ctr - wh2, ctr + wh2], dim=2)
    scores = torch.rand(B, N)
    for conf in (0.0, 0.35):
        want_idx, want_cnt = eager.nms_batched(boxes, scores, 0.5, conf)
        got_idx, got_cnt = _C().nms_batched(boxes.cuda(), scores.cuda(),
                                            0.5, conf)
        assert got_cnt.cpu().tolist() == want_cnt.tolist()
        for i in range(B):
            k = int(want_cnt[i])
            assert got_idx[i, :k].cpu().tolist() == \
                want_idx[i, :k].tolist(), (conf, i)


def test_decode_matches_eager():
    from real_time_helmet_detection_amd.ops import eager
    torch.manual_seed(6)
    b, c, h, w = 3, 2, 64, 64
    hm = torch.sigmoid(torch.randn(b, c, h, w) * 3)
    off = torch.rand(b, 2, h, w)
    wh = torch.rand(b, 2, h, w) * 20
    want_boxes, want_cls, want_scores = eager.batched_decode(
        hm, off, wh, 4, 50, 3, False)
    boxes, clss, scores = _C().decode_fwd(hm.cuda(), off.cuda(), wh.cuda(),
                                          4, 50, 3, False)
    # order may differ on exact ties; compare sorted by score then x1
    for i in range(b):
        ws, wo = want_scores[i].sort(descending=True)
        gs, go = scores[i].cpu().sort(descending=True)
        torch.testing.assert_close(gs, ws, rtol=1e-5, atol=1e-6)
        keep = ws > 0
        torch.testing.assert_close(
            boxes[i].cpu()[go][keep], want_boxes[i][wo][keep],
            rtol=1e-4, atol=1e-4)
        assert (clss[i].cpu()[go][keep] == want_cls[i][wo][keep]).all()


def test_nms_matches_eager():
    from real_time_helmet_detection_amd.ops import eager
    torch.manual_seed(7)
    xy = torch.rand(200, 2) * 100
    wh = torch.rand(200, 2) * 30 + 1
    boxes = torch.cat([xy, xy + wh], dim=1)
    scores = torch.rand(200)
    want = eager.nms(boxes, scores, 0.5)
    got = _C().nms_fwd(boxes.cuda(), scores.cuda(), 0.5)
    assert got.cpu().tolist() == want.tolist()


# ------------------------------------------------------------------- conv --

@pytest.mark.parametrize('cfg', [
    # (Cin, Cout, k, stride, pad, H)
    (32, 32, 3, 1, 1, 16),
    (128, 128, 3, 1, 1, 16),
    (64, 128, 1, 1, 0, 16),
    (128, 6, 1, 1, 0, 16),
    (128, 128, 2, 2, 0, 16),
    (96, 160, 3, 1, 1, 12),   # non-pow2 channels, odd spatial
])
def test_conv_fwd_f32_exact(cfg):
    cin, cout, k, stride, pad, h = cfg
    torch.manual_seed(8)
    x = torch.randn(2, cin, h, h)
    w = torch.randn(cout, cin, k, k) * 0.1
    bias = torch.randn(cout)
    want = F.conv2d(x, w, bias, stride=stride, padding=pad)

    wpk = _C().pack_weights(w.cuda(), False, False)
    ones = torch.ones(cout, device='cuda')
    got = _C().conv_fwd(to_gpu(x), wpk, ones, bias.cuda(), None, k, k,
                        stride, pad, cout, 0)
    assert rel_err(got, want) < 1e-5


def test_conv_fwd_bf16_close():
    torch.manual_seed(9)
    x = torch.randn(2, 128, 16, 16)
    w = torch.randn(128, 128, 3, 3) * 0.05
    want = F.conv2d(x.to(torch.bfloat16).float(),
                    w.to(torch.bfloat16).float(), None, padding=1)
    wpk = _C().pack_weights(w.cuda(), False, True)
    ones = torch.ones(128, device='cuda')
    zeros = torch.zeros(128, device='cuda')
    got = _C().conv_fwd(to_gpu(x, torch.bfloat16), wpk, ones, zeros, None,
                        3, 3, 1, 1, 128, 0)
    assert rel_err(got, want) < 0.03


def test_conv_fwd_relu_epilogue():
    torch.manual_seed(10)
    x = torch.randn(1, 32, 8, 8)
    w = torch.randn(32, 32, 3, 3) * 0.1
    want = F.relu(F.conv2d(x, w, None, padding=1) * 2.0 + 0.5)
    wpk = _C().pack_weights(w.cuda(), False, False)
    sc = torch.full((32,), 2.0, device='cuda')
    sh = torch.full((32,), 0.5, device='cuda')
    got = _C().conv_fwd(to_gpu(x), wpk, sc, sh, None, 3, 3, 1, 1, 32, 1)
    assert rel_err(got, want) < 1e-5


@pytest.mark.parametrize('cfg', [
    # (B, Cin, Cout, k, pad, H, splitk)
    (16, 128, 128, 3, 1, 8, 1),
    (16, 128, 128, 3, 1, 8, 4),
    (16, 128, 128, 3, 1, 16, 8),
    (16, 128, 128, 3, 1, 32, 16),
    (2, 128, 6, 1, 0, 16, 2),     # head shape: Cout not 64-aligned
    (3, 96, 160, 3, 1, 12, 4),    # ragged M, Cout>64 non-mult
])
def test_conv_fwd_small_splitk(cfg):
    """64x64-tile (+split-K) variant vs torch conv on bf16-rounded data,
    including the fused scale/shift/relu(+skip) epilogue."""
    b, cin, cout, k, pad, h, sk = cfg
    torch.manual_seed(13)
    x = torch.randn(b, cin, h, h).to(torch.bfloat16).float()
    w = (torch.randn(cout, cin, k, k) * 0.05).to(torch.bfloat16).float()
    skip = torch.randn(b, cout, h, h).to(torch.bfloat16).float()
    sc = torch.rand(cout) + 0.5
    sh = torch.randn(cout) * 0.1
    want = F.relu(F.conv2d(x, w, None, padding=pad)
                  * sc.view(1, -1, 1, 1) + sh.view(1, -1, 1, 1) + skip)

    wpk = _C().pack_weights(w.cuda(), False, True)
    got = _C().conv_fwd_small(to_gpu(x, torch.bfloat16), wpk, sc.cuda(),
                              sh.cuda(), to_gpu(skip, torch.bfloat16),
                              k, k, 1, pad, cout, 1, sk)
    assert rel_err(got, want) < 0.03


@pytest.mark.parametrize('cfg', [
    # (B, Cin, Cout, k, pad, H)
    (4, 128, 128, 3, 1, 32),
    (2, 64, 128, 3, 1, 24),      # Cin=64 (one zero K-half)
    (2, 128, 6, 1, 0, 16),       # head
    (3, 96, 160, 3, 1, 12),      # ragged everything
])
def test_conv_fwd_k64(cfg):
    """64-ch K-step variant vs torch conv on bf16-rounded data."""
    b, cin, cout, k, pad, h = cfg
    torch.manual_seed(23)
    x = torch.randn(b, cin, h, h).to(torch.bfloat16).float()
    w = (torch.randn(cout, cin, k, k) * 0.05).to(torch.bfloat16).float()
    skip = torch.randn(b, cout, h, h).to(torch.bfloat16).float()
    sc = torch.rand(cout) + 0.5
    sh = torch.randn(cout) * 0.1
    want = F.relu(F.conv2d(x, w, None, padding=pad)
                  * sc.view(1, -1, 1, 1) + sh.view(1, -1, 1, 1) + skip)
    wpk = _C().pack_weights(w.cuda(), False, True)
    got = _C().conv_fwd_k64(to_gpu(x, torch.bfloat16), wpk, sc.cuda(),
                            sh.cuda(), to_gpu(skip, torch.bfloat16),
                            k, k, 1, pad, cout, 1)
    assert rel_err(got, want) < 0.03


@pytest.mark.parametrize('hw', [16, 48])
def test_conv_fwd_stats_matches_bn_stats(hw):
    """Fused-epilogue BN stats == standalone bn_stats on the conv output
    (and the conv output itself matches conv_fwd)."""
    torch.manual_seed(24)
    x = torch.randn(3, 32, hw, hw).to(torch.bfloat16).float()
    w = (torch.randn(32, 32, 3, 3) * 0.1).to(torch.bfloat16).float()
    bias = torch.randn(32)
    wpk = _C().pack_weights(w.cuda(), False, True)
    ones = torch.ones(32, device='cuda')
    y_ref = _C().conv_fwd(to_gpu(x, torch.bfloat16), wpk, ones,
                          bias.cuda(), None, 3, 3, 1, 1, 32, 0)
    outs = _C().conv_fwd_stats(to_gpu(x, torch.bfloat16), wpk, ones,
                               bias.cuda(), 3, 3, 1, 1, 32, 0)
    assert rel_err(outs[0], y_ref) < 1e-6
    want_mean, want_rstd = _C().bn_stats(y_ref, None, None, 0.1, 1e-5)
    if len(outs) == 3:
        rm = torch.zeros(32, device='cuda')
        rv = torch.ones(32, device='cuda')
        mean, rstd = _C().bn_stats_from_parts(outs[1], outs[2], rm, rv,
                                              0.1, 1e-5,
                                              y_ref.numel() // 32)
        assert rel_err(mean, want_mean) < 1e-4
        assert rel_err(rstd, want_rstd) < 1e-4
        # running stats updated like bn_stats does
        rm2 = torch.zeros(32, device='cuda')
        rv2 = torch.ones(32, device='cuda')
        _C().bn_stats(y_ref, rm2, rv2, 0.1, 1e-5)
        assert rel_err(rm, rm2) < 1e-4
        assert rel_err(rv, rv2) < 1e-4


def test_conv_autotune_dispatch_matches_big():
    """The autotuned conv_fwd on a small-M shape must agree with the
    explicit variants (whichever the cache picked)."""
    torch.manual_seed(14)
    x = torch.randn(16, 128, 8, 8).to(torch.bfloat16).float()
    w = (torch.randn(128, 128, 3, 3) * 0.05).to(torch.bfloat16).float()
    want = F.conv2d(x, w, None, padding=1)
    wpk = _C().pack_weights(w.cuda(), False, True)
    ones = torch.ones(128, device='cuda')
    zeros = torch.zeros(128, device='cuda')
    got = _C().conv_fwd(to_gpu(x, torch.bfloat16), wpk, ones, zeros, None,
                        3, 3, 1, 1, 128, 0)
    assert rel_err(got, want) < 0.03
    # second call takes the cached choice — still correct
    got2 = _C().conv_fwd(to_gpu(x, torch.bfloat16), wpk, ones, zeros, None,
                         3, 3, 1, 1, 128, 0)
    assert rel_err(got2, want) < 0.03


def test_conv_fwd_fp8_multi_kblock():
    """Cin=256 (increase-ch configs): two 128-k blocks per tap — the
    kc>1 staging walk of the fp8 kernel."""
    torch.manual_seed(27)
    x8 = (torch.randn(2, 256, 12, 12) * 0.5).to(torch.float8_e4m3fn)
    w = torch.randn(128, 256, 3, 3) * 0.05
    sw = w.abs().amax(dim=(1, 2, 3)).clamp(min=1e-8) / 240.0
    w8 = (w / sw.view(-1, 1, 1, 1)).to(torch.float8_e4m3fn).float() \
        * sw.view(-1, 1, 1, 1)
    want = F.conv2d(x8.float(), w8, None, padding=1)
    wpk = _C().pack_weights_fp8((w / sw.view(-1, 1, 1, 1)).cuda())
    got = _C().conv_fwd_fp8r(
        x8.cuda().contiguous(memory_format=CL), wpk, sw.cuda(),
        torch.zeros(128, device='cuda'), None, 3, 3, 1, 1, 128, 0, False)
    assert rel_err(got.float(), want) < 0.05


@pytest.mark.parametrize('out_fp8', [True, False])
def test_conv_fwd_fp8_resident(out_fp8):
    """fp8-resident conv (e4m3 in, e4m3/bf16 out, fused epilogue + fp8
    skip) vs torch conv on the dequantized values."""
    torch.manual_seed(21)
    x8 = (torch.randn(4, 32, 16, 16) * 0.5).to(torch.float8_e4m3fn)
    x = x8.float()
    w = torch.randn(32, 32, 3, 3) * 0.1
    sw = w.abs().amax(dim=(1, 2, 3)).clamp(min=1e-8) / 240.0
    w8 = (w / sw.view(-1, 1, 1, 1)).to(torch.float8_e4m3fn).float() \
        * sw.view(-1, 1, 1, 1)
    sk8 = (torch.randn(4, 32, 16, 16) * 0.5).to(torch.float8_e4m3fn)
    sc = torch.rand(32) + 0.5
    sh = torch.randn(32) * 0.1
    want = F.relu(F.conv2d(x, w8, None, padding=1) * sc.view(1, -1, 1, 1)
                  + sh.view(1, -1, 1, 1) + sk8.float())

    wpk = _C().pack_weights_fp8((w / sw.view(-1, 1, 1, 1)).cuda())
    got = _C().conv_fwd_fp8r(
        x8.cuda().contiguous(memory_format=CL), wpk,
        (sc * sw).cuda(), sh.cuda(),
        sk8.cuda().contiguous(memory_format=CL),
        3, 3, 1, 1, 32, 1, out_fp8)
    assert got.dtype == (torch.float8_e4m3fn if out_fp8
                         else torch.bfloat16)
    assert rel_err(got.float(), want) < (0.08 if out_fp8 else 0.05)


def test_fp8_elementwise_ops():
    """fp8 instantiations of pool/upsample/add (the fp8-resident chain's
    glue ops) vs f32 on dequantized values."""
    torch.manual_seed(22)
    x8 = (torch.randn(2, 16, 8, 8) * 2).to(torch.float8_e4m3fn)
    x = x8.float()
    g8 = x8.cuda().contiguous(memory_format=CL)

    got = _C().pool2x2_fwd(g8, True, False)[0]
    assert got.dtype == torch.float8_e4m3fn
    assert rel_err(got.float(), F.max_pool2d(x, 2, 2)) < 1e-6  # exact copy

    sk8 = (torch.randn(2, 16, 16, 16)).to(torch.float8_e4m3fn)
    got = _C().upsample2x_add_fwd(g8, sk8.cuda().contiguous(
        memory_format=CL))
    want = F.interpolate(x, scale_factor=2, mode='nearest') + sk8.float()
    assert rel_err(got.float(), want) < 0.08

    b8 = (torch.randn(2, 16, 8, 8)).to(torch.float8_e4m3fn)
    got = _C().add_act_fwd(g8, b8.cuda().contiguous(memory_format=CL), 1)
    want = F.relu(x + b8.float())
    assert rel_err(got.float(), want) < 0.08


def test_dgrad_via_swapped_pack():
    # conv3x3 s1 p1: dX = conv(dY, rot180(W) transposed)
    torch.manual_seed(11)
    x = torch.randn(2, 64, 12, 12).requires_grad_(True)
    w = torch.randn(32, 64, 3, 3) * 0.1
    y = F.conv2d(x, w, None, padding=1)
    dy = torch.randn_like(y)
    y.backward(dy)
    wpk_t = _C().pack_weights(w.cuda(), True, False)
    ones = torch.ones(64, device='cuda')
    zeros = torch.zeros(64, device='cuda')
    got_dx = _C().conv_fwd(to_gpu(dy.detach()), wpk_t, ones, zeros, None,
                           3, 3, 1, 1, 64, 0)
    assert rel_err(got_dx, x.grad) < 1e-5


@pytest.mark.parametrize('cfg', [
    (64, 32, 3, 1, 1, 12),
    (128, 128, 1, 1, 0, 16),
    (3, 64, 7, 2, 3, 32),    # stem wgrad
])
def test_wgrad(cfg):
    cin, cout, k, stride, pad, h = cfg
    torch.manual_seed(12)
    x = torch.randn(2, cin, h, h)
    w = torch.randn(cout, cin, k, k, requires_grad=True)
    y = F.conv2d(x, w, None, stride=stride, padding=pad)
    dy = torch.randn_like(y)
    y.backward(dy)
    got = _C().wgrad(to_gpu(x), to_gpu(dy.detach()), k, k, stride, pad)
    assert rel_err(got, w.grad) < 1e-4


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_stem_im2col_conv(dtype):
    """Unified stem path (any dtype): unfold + 1x1 MFMA conv == F.conv2d."""
    torch.manual_seed(13)
    x = torch.randn(2, 3, 64, 64)
    w = torch.randn(64, 3, 7, 7) * 0.1
    bias = torch.randn(64)
    if dtype == torch.bfloat16:
        x = x.to(dtype).float()
        w = w.to(dtype).float()
    want = F.conv2d(x, w, bias, stride=2, padding=3)

    from real_time_helmet_detection_amd.ops.hip import _stem_col_weight
    xcol = _C().stem_im2col(to_gpu(x, dtype), 7, 2, 3)
    assert xcol.shape[1] == 152
    wpk = _C().pack_weights(_stem_col_weight(w.cuda()), False,
                            dtype == torch.bfloat16)
    ones = torch.ones(64, device='cuda')
    got = _C().conv_fwd(xcol, wpk, ones, bias.cuda(), None, 1, 1, 1, 0,
                        64, 0)
    tol = 0.02 if dtype == torch.bfloat16 else 1e-5
    assert rel_err(got, want) < tol


# -------------------------------------------------------------------- bn ---

def test_bn_stats_and_fwd():
    torch.manual_seed(14)
    x = torch.randn(4, 32, 8, 8) * 3 + 1
    rm = torch.zeros(32)
    rv = torch.ones(32)
    bn = torch.nn.BatchNorm2d(32)
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-0.5, 0.5)
    want = F.relu(F.batch_norm(x, rm.clone(), rv.clone(), bn.weight, bn.bias,
                               training=True, momentum=0.1, eps=1e-5))

    rm_g, rv_g = rm.clone().cuda(), rv.clone().cuda()
    mean, rstd = _C().bn_stats(to_gpu(x), rm_g, rv_g, 0.1, 1e-5)
    got = _C().bn_act_fwd(to_gpu(x), mean, rstd,
                          bn.weight.detach().cuda(),
                          bn.bias.detach().cuda(), 1)
    assert rel_err(got, want) < 1e-4
    # running stats updated torch-style
    rm_ref = rm.clone()
    rv_ref = rv.clone()
    F.batch_norm(x, rm_ref, rv_ref, bn.weight, bn.bias, training=True,
                 momentum=0.1, eps=1e-5)
    assert rel_err(rm_g, rm_ref) < 1e-4
    assert rel_err(rv_g, rv_ref) < 1e-4


def test_bn_act_bwd():
    torch.manual_seed(15)
    x = torch.randn(4, 32, 8, 8, requires_grad=True)
    bn = torch.nn.BatchNorm2d(32)
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-0.5, 0.5)
    y = F.relu(bn(x))
    dy = torch.randn_like(y)
    y.backward(dy)

    mean, rstd = _C().bn_stats(to_gpu(x.detach()), None, None, 0.1, 1e-5)
    dx, dgamma, dbeta = _C().bn_act_bwd(to_gpu(dy), to_gpu(x.detach()),
                                        mean, rstd,
                                        bn.weight.detach().cuda(),
                                        bn.bias.detach().cuda(), 1)
    assert rel_err(dx, x.grad) < 1e-3
    assert rel_err(dgamma, bn.weight.grad) < 1e-3
    assert rel_err(dbeta, bn.bias.grad) < 1e-3


def test_col_sum():
    torch.manual_seed(16)
    x = torch.randn(3, 33, 7, 9)  # odd sizes
    want = x.sum(dim=(0, 2, 3))
    got = _C().col_sum(to_gpu(x))
    assert rel_err(got, want) < 1e-4


@pytest.mark.parametrize('cin', [6, 3, 20])
def test_conv_fwd_small_cin(cin):
    """Cin not a multiple of 32/8 (head dgrad, merge_prediction convs)."""
    torch.manual_seed(17)
    x = torch.randn(2, cin, 12, 12)
    w = torch.randn(64, cin, 3, 3) * 0.1
    want = F.conv2d(x, w, None, padding=1)
    wpk = _C().pack_weights(w.cuda(), False, False)
    ones = torch.ones(64, device='cuda')
    zeros = torch.zeros(64, device='cuda')
    got = _C().conv_fwd(to_gpu(x), wpk, ones, zeros, None, 3, 3, 1, 1, 64, 0)
    assert rel_err(got, want) < 1e-5


@pytest.mark.parametrize('cfg', [
    (64, 32, 3, 1, 1, 12),
    (128, 128, 3, 1, 1, 16),
    (128, 128, 1, 1, 0, 16),
    (3, 64, 7, 2, 3, 32),
    (128, 6, 1, 1, 0, 16),
])
def test_wgrad_bf16_fast(cfg):
    cin, cout, k, stride, pad, h = cfg
    torch.manual_seed(18)
    x = (torch.randn(2, cin, h, h)).to(torch.bfloat16).float()
    w = torch.randn(cout, cin, k, k, requires_grad=True)
    y = F.conv2d(x, w, None, stride=stride, padding=pad)
    dy = torch.randn_like(y).to(torch.bfloat16).float()
    y.backward(dy)
    got = _C().wgrad_bf16_fast(to_gpu(x, torch.bfloat16),
                               to_gpu(dy, torch.bfloat16), k, k, stride, pad)
    assert rel_err(got, w.grad) < 0.03


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_stem_wgrad_im2col(dtype):
    """Stem wgrad through the unfolded tensor (both engines)."""
    torch.manual_seed(19)
    x = torch.randn(2, 3, 64, 64)
    if dtype == torch.bfloat16:
        x = x.to(dtype).float()
    w = torch.randn(64, 3, 7, 7, requires_grad=True)
    y = F.conv2d(x, w, None, stride=2, padding=3)
    dy = torch.randn_like(y)
    if dtype == torch.bfloat16:
        dy = dy.to(dtype).float()
    y.backward(dy)

    xcol = _C().stem_im2col(to_gpu(x, dtype), 7, 2, 3)
    if dtype == torch.bfloat16:
        dwc = _C().wgrad_bf16_fast(xcol, to_gpu(dy, dtype), 1, 1, 1, 0)
    else:
        dwc = _C().wgrad(xcol, to_gpu(dy, dtype), 1, 1, 1, 0)
    got = (dwc[:, :49 * 3, 0, 0].reshape(64, 49, 3).permute(0, 2, 1)
           .reshape(64, 3, 7, 7).contiguous())
    tol = 0.02 if dtype == torch.bfloat16 else 1e-4
    assert rel_err(got, w.grad) < tol


@pytest.mark.gpu
def test_wgrad_deterministic():
    """Per-chunk-partial writeback: identical inputs must give BITWISE
    identical weight gradients across runs (the atomicAdd version varied
    with fp add order)."""
    from real_time_helmet_detection_amd.ops import _backend
    C = _backend.require_ext()
    torch.manual_seed(0)
    x = torch.randn(4, 64, 32, 32, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = torch.randn(4, 64, 32, 32, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    a = C.wgrad_bf16_fast(x, dy, 3, 3, 1, 1)
    b = C.wgrad_bf16_fast(x, dy, 3, 3, 1, 1)
    assert torch.equal(a, b)
