"""GPU execution path: autograd wrappers over the gfx950 kernels (_C).

Every function here matches an eager twin in ``ops.eager`` /
``ops.functional`` — same signatures, same numerics (tested in
tests/test_gpu_kernels.py against fp32 eager oracles).

Structure per op: a torch.autograd.Function whose forward/backward call the
HIP kernels directly. Convolutions:

- forward: pack_weights (torch (Cout,Cin,KH,KW) fp32 master -> MFMA-friendly
  [taps][Coutp][Cinp] bf16/f32) -> conv_fwd implicit GEMM with fused
  scale/shift/act epilogue. Training-mode BN splits into conv(linear) ->
  bn_stats -> bn_act_fwd; inference folds BN into the epilogue.
- backward: dgrad = conv_fwd on dY with rotated/transposed packed weights;
  wgrad = the transposed-staging MFMA kernel; BN/bias grads from the
  reduction kernels.

bf16 policy: when rthd.amp autocast is active (or inputs are bf16) compute
runs on the bf16 MFMA pipe with fp32 accumulate; otherwise the exact-f32
MFMA path (guide §3: v_mfma_f32_16x16x4_f32 — identical numerics to an fmaf
chain).
"""

import os

import torch

from . import _backend
from .. import amp as _amp

ACT_CODE = {'Linear': 0, 'ReLU': 1, 'LReLU': 2}

# side stream for weight-gradient GEMMs: wgrad and dgrad both consume the
# same upstream gradient and are independent, so wgrad runs concurrently
# with the dgrad/BN chain (fills the CU idle time of the small layers).
_wgrad_streams = {}


def _wgrad_stream(device):
    st = _wgrad_streams.get(device)
    if st is None:
        st = torch.cuda.Stream(device=device)
        _wgrad_streams[device] = st
    return st


def _C():
    return _backend.require_ext()


def _ops():
    """torch.ops.rthd — the dispatcher-registered views of the inference-path
    kernels (TORCH_LIBRARY in bindings.cpp). Used for every forward call so
    torch.jit.trace records the native ops and the exported GPU model keeps
    the gfx950 kernels (reference export.py:120-130's portability contract,
    upgraded to native)."""
    _backend.require_ext()
    return torch.ops.rthd


def _bf16_mode(x):
    return x.dtype == torch.bfloat16 or _amp.is_autocast_enabled()


# Monotone stamp bumped on every training-mode conv forward (and on every
# hipGraph replay of a captured training step, where no python runs): BN
# running stats are updated by raw kernel writes that never bump
# tensor._version, so the folded-BN inference caches key on this stamp to
# stay fresh across train->eval->train transitions.
_train_stamp = [0]


def bump_train_stamp():
    _train_stamp[0] += 1


# per-(device, n) constant epilogue vectors: the training path needs
# ones/zeros per conv call — allocating them fresh was ~130 tiny kernel
# launches per train step
_const_vecs = {}


def _ones_zeros(n, device):
    key = (device.index if device.index is not None else -1, n)
    v = _const_vecs.get(key)
    if v is None:
        v = (torch.ones(n, device=device, dtype=torch.float32),
             torch.zeros(n, device=device, dtype=torch.float32))
        _const_vecs[key] = v
    return v


# ------------------------------------------------------------------ conv ---

def _stem_col_weight(weight):
    """(Cout,3,KS,KS) -> (Cout, KS*KS*3 padded to x8, 1, 1): column t*3+ci,
    matching stem_im2col's unfolded layout (7x7: 147 -> 152 cols)."""
    cout = weight.shape[0]
    w = weight.permute(0, 2, 3, 1).reshape(cout, -1)
    padc = (-w.shape[1]) % 8
    if padc:
        w = torch.nn.functional.pad(w, (0, padc))
    return w.reshape(cout, -1, 1, 1)


class _ConvBNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, gamma, beta, rmean, rvar, skip,
                kh, kw, stride, pad, act_code, use_bn, training, momentum,
                eps, is_stem):
        C = _C()
        # the bf16 glds staging needs 16-B channel chunks; rare non-multiple
        # channel counts (merge_prediction 6->128) take the exact-f32 path.
        # The stem has its own direct kernel (any Cin) and stays bf16.
        bf16 = _bf16_mode(x) and (is_stem or x.shape[1] % 8 == 0)
        dtype = torch.bfloat16 if bf16 else torch.float32
        xc = x.to(dtype).contiguous(memory_format=torch.channels_last)
        skc = None
        if skip is not None:
            skc = skip.to(dtype).contiguous(
                memory_format=torch.channels_last)
        cout = weight.shape[0]
        dev = x.device

        # stem (any dtype): unfold to [px][152] once and run the MFMA conv
        # as its 1x1 case — faster than a direct VALU stem kernel AND the
        # unfolded tensor is reused by the backward wgrad (which otherwise
        # re-unfolds).
        stem_col = is_stem and kh == 7
        if stem_col:
            xcol = C.stem_im2col(xc, kh, stride, pad)
            wpk = C.pack_weights(_stem_col_weight(weight), False, bf16)
            xc = xcol
        else:
            wpk = C.pack_weights(weight, False, bf16)

        ones, zeros = _ones_zeros(cout, dev)

        ops = _ops()

        def run_conv(scale, shift, act, sk=None):
            if stem_col:
                assert sk is None
                return ops.conv_fwd(xc, wpk, scale, shift, None, 1, 1, 1,
                                    0, cout, act)
            return ops.conv_fwd(xc, wpk, scale, shift, sk, kh, kw, stride,
                                pad, cout, act)

        bias_f = bias.float().contiguous() if bias is not None else zeros
        mean = rstd = y_lin = None
        if use_bn and training:
            # conv output INCLUDES the (redundant-under-BN) bias so running
            # stats match the eager/reference semantics (stem has bias+BN).
            # Fused-epilogue BN stats exist (conv_fwd_stats) but measured
            # ~6% SLOWER end to end than the standalone colsum pass (the
            # extra epilogue stores + shuffles extend the conv's
            # store-tail more than the saved 0.4 ms read pass; same-box
            # A/B 940 vs 997 img/s) — opt-in via RTHD_FUSED_STATS=1.
            if bf16 and os.environ.get('RTHD_FUSED_STATS') == '1':
                if stem_col:
                    outs = C.conv_fwd_stats(xc, wpk, ones, bias_f, 1, 1,
                                            1, 0, cout, ACT_CODE['Linear'])
                else:
                    outs = C.conv_fwd_stats(xc, wpk, ones, bias_f, kh, kw,
                                            stride, pad, cout,
                                            ACT_CODE['Linear'])
                y_lin = outs[0]
                if len(outs) == 3:
                    mean, rstd = C.bn_stats_from_parts(
                        outs[1], outs[2], rmean, rvar, momentum, eps,
                        y_lin.numel() // cout)
                else:
                    mean, rstd = C.bn_stats(y_lin, rmean, rvar, momentum,
                                            eps)
            else:
                y_lin = run_conv(ones, bias_f, ACT_CODE['Linear'])
                mean, rstd = C.bn_stats(y_lin, rmean, rvar, momentum, eps)
            y = C.bn_act_fwd(y_lin, mean, rstd, gamma, beta, act_code, skc)
        elif use_bn:
            rstd_run = torch.rsqrt(rvar.float() + eps)
            scale = (gamma.float() * rstd_run).contiguous()
            shift = (beta.float() + (bias_f - rmean.float()) * scale
                     ).contiguous()
            y = run_conv(scale, shift, act_code, skc)
        else:
            y = run_conv(ones, bias_f, act_code, skc)

        ctx.meta = (kh, kw, stride, pad, act_code, use_bn, training, bf16,
                    is_stem, bias is not None, eps, skc is not None,
                    stem_col)
        if use_bn and training:
            ctx.save_for_backward(xc, weight, gamma, beta, y_lin, mean,
                                  rstd, skc)
        elif use_bn:
            ctx.save_for_backward(xc, weight, gamma, beta, y,
                                  rmean.detach().clone(),
                                  rvar.detach().clone())
        else:
            ctx.save_for_backward(xc, weight, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _C()
        (kh, kw, stride, pad, act_code, use_bn, training, bf16, is_stem,
         has_bias, eps, has_skip, stem_col) = ctx.meta
        dgamma = dbeta = dbias = dskip = None

        if use_bn and training:
            (xc, weight, gamma, beta, y_lin, mean, rstd,
             skc) = ctx.saved_tensors
            outs = C.bn_act_bwd(dy, y_lin, mean, rstd, gamma, beta,
                                act_code, skc)
            if has_skip:
                dpre, dgamma, dbeta, dskip = outs
            else:
                dpre, dgamma, dbeta = outs
        elif use_bn:
            # eval-mode BN backward (rare: grads through a frozen BN)
            xc, weight, gamma, beta, y, rmean, rvar = ctx.saved_tensors
            dact = C.add_act_bwd(dy, y, act_code)
            if has_skip:
                dskip = dact
            scale = (gamma.float() * torch.rsqrt(rvar.float() + eps))
            dpre = (dact.float() * scale.view(1, -1, 1, 1)).to(dact.dtype)
            dpre = dpre.contiguous(memory_format=torch.channels_last)
        else:
            xc, weight, y = ctx.saved_tensors
            dpre = C.add_act_bwd(dy, y, act_code)
            if has_skip:
                dskip = dpre
        if has_bias and ctx.needs_input_grad[2]:
            if use_bn and training:
                # train-mode BN re-centers the conv output, so a constant
                # bias shift cancels exactly: dL/dbias = sum(dpre) = 0
                # analytically (sum of BN-backward input grads over the
                # normalization axes is 0). Skip the reduction.
                dbias = torch.zeros(dpre.shape[1], device=dpre.device,
                                    dtype=torch.float32)
            else:
                dbias = C.col_sum(dpre)

        # wgrad — on the side stream, overlapping the dgrad launched above
        dw = None
        if ctx.needs_input_grad[1]:
            cur = torch.cuda.current_stream(xc.device)
            side = _wgrad_stream(xc.device)
            side.wait_stream(cur)
            with torch.cuda.stream(side):
                if stem_col:
                    # xc is the SAVED unfolded [px][152] tensor from the
                    # forward: the wgrad is its 1x1 MFMA case; column
                    # t*3+ci -> dW[co][ci][t]
                    cout = dpre.shape[1]
                    if xc.dtype == torch.bfloat16:
                        dwc = C.wgrad_bf16_fast(xc, dpre, 1, 1, 1, 0)
                    else:
                        dwc = C.wgrad(xc, dpre.float(), 1, 1, 1, 0)
                    dw = (dwc[:, :kh * kw * 3, 0, 0]
                          .reshape(cout, kh * kw, 3).permute(0, 2, 1)
                          .reshape(cout, 3, kh, kw).contiguous())
                elif xc.dtype == torch.bfloat16:
                    # pad stray non-x8 channel counts (head Cout=6) so the
                    # aligned fast kernel runs; slice the result back
                    cin_p = xc.shape[1] % 8
                    cout_p = dpre.shape[1] % 8
                    xs = torch.nn.functional.pad(
                        xc, (0, 0, 0, 0, 0, 8 - cin_p)).contiguous(
                        memory_format=torch.channels_last) if cin_p else xc
                    ds = torch.nn.functional.pad(
                        dpre, (0, 0, 0, 0, 0, 8 - cout_p)).contiguous(
                        memory_format=torch.channels_last) if cout_p                         else dpre
                    dw = C.wgrad_bf16_fast(xs, ds, kh, kw, stride, pad)
                    if cin_p or cout_p:
                        dw = dw[:dpre.shape[1], :xc.shape[1]].contiguous(
                            memory_format=torch.channels_last)
                else:
                    dw = C.wgrad(xc, dpre, kh, kw, stride, pad)
            cur.wait_stream(side)
            dw.record_stream(cur)

        # dgrad
        dx = None
        if ctx.needs_input_grad[0]:
            if is_stem:
                raise NotImplementedError(
                    'stem dgrad (3-channel input) is not needed: the image '
                    'is a leaf tensor')
            cin = xc.shape[1]
            ones, zeros = _ones_zeros(cin, xc.device)
            if stride == 1:
                bf16_d = bf16 and dpre.shape[1] % 8 == 0
                wpk_t = C.pack_weights(weight, True, bf16_d)
                dsrc = dpre if bf16_d else dpre.float()
                dx = C.conv_fwd(dsrc, wpk_t, ones, zeros, None, kh, kw, 1,
                                pad, cin, ACT_CODE['Linear'])
                if dx.dtype != xc.dtype:
                    dx = dx.to(xc.dtype)
            elif stride == 2 and kh == 2 and kw == 2 and pad == 0:
                # k2/s2 windows don't overlap: each input pixel receives
                # from exactly one output pixel, so dgrad decomposes into
                # FOUR 1x1 convs (one per tap parity), scattered back into
                # the interleaved input grid (the pool='Conv' downsampler).
                dx = torch.empty_like(xc).contiguous(
                    memory_format=torch.channels_last)
                for r in range(2):
                    for c in range(2):
                        wt = weight[:, :, r:r + 1, c:c + 1]
                        wpk_t = C.pack_weights(wt, True, bf16)
                        part = C.conv_fwd(dpre, wpk_t, ones, zeros, None,
                                          1, 1, 1, 0, cin,
                                          ACT_CODE['Linear'])
                        dx[:, :, r::2, c::2] = part
            else:
                raise NotImplementedError(
                    f'HIP dgrad for k={kh} stride={stride} not implemented')

        return (dx, dw, dbias, dgamma, dbeta, None, None, dskip,
                None, None, None, None, None, None, None, None, None, None)


def _conv_infer_fp8(x, conv, bn, act_code, skip=None):
    """fp8-RESIDENT inference path (BASELINE config 5): activations stay
    e4m3 between layers. x arrives bf16 only at the stem boundary (one
    cast); weights are per-cout-scaled e4m3 with the scale folded into the
    f32 epilogue (cached on the module, keyed by weight version + train
    stamp); output stays e4m3 for convs with BN (mid-network) and drops to
    bf16 at the heads that feed the decode."""
    C = _C()
    kh, kw = conv.kernel_size
    stride, pad = conv.stride[0], conv.padding[0]
    cout = conv.weight.shape[0]
    cache = getattr(conv, '_rthd_fp8_cache', None)
    ver = (conv.weight._version, _train_stamp[0])
    if cache is None or cache[0] != ver:
        w = conv.weight.detach().float()
        sw = w.abs().amax(dim=(1, 2, 3)).clamp(min=1e-8) / 240.0
        wpk = C.pack_weights_fp8(w / sw.view(-1, 1, 1, 1))
        bias_f = (conv.bias.float() if conv.bias is not None
                  else torch.zeros_like(sw))
        if bn is not None:
            rstd_run = torch.rsqrt(bn.running_var.float() + bn.eps)
            scale = (bn.weight.float() * rstd_run) * sw
            shift = (bn.bias.float()
                     + (bias_f - bn.running_mean.float())
                     * bn.weight.float() * rstd_run)
        else:
            scale = sw
            shift = bias_f
        conv._rthd_fp8_cache = (ver, wpk, scale.contiguous(),
                                shift.contiguous())
        cache = conv._rthd_fp8_cache
    _, wpk, scale, shift = cache

    if x.dtype != torch.float8_e4m3fn:
        x = x.to(torch.float8_e4m3fn)
    xc = x.contiguous(memory_format=torch.channels_last)
    skc = None
    if skip is not None:
        skc = skip if skip.dtype == torch.float8_e4m3fn \
            else skip.to(torch.float8_e4m3fn)
        skc = skc.contiguous(memory_format=torch.channels_last)
    out_fp8 = bn is not None  # heads (no BN) emit bf16 for the decode
    return _ops().conv_fwd_fp8r(xc, wpk, scale, shift, skc, kh, kw,
                                stride, pad, cout, act_code, out_fp8)


def _conv_infer(x, conv, bn, act_code, skip, kh, kw, stride, pad, is_stem):
    """Plain (non-autograd.Function) inference forward: eval-BN fold into
    the conv epilogue via torch.ops.rthd.* only — no PythonOp nodes, so
    torch.jit.trace serializes it (export.py's native GPU trace).

    Packed weights + folded scale/shift are cached on the conv module keyed
    by the parameter/buffer versions: the serving hot path skips the
    per-call pack + fold, and a trace bakes the frozen tensors as
    constants."""
    ops = _ops()
    bf16 = _bf16_mode(x) and (is_stem or x.shape[1] % 8 == 0)
    dtype = torch.bfloat16 if bf16 else torch.float32
    xc = x.to(dtype).contiguous(memory_format=torch.channels_last)
    skc = None
    if skip is not None:
        skc = skip.to(dtype).contiguous(memory_format=torch.channels_last)
    weight = conv.weight
    cout = weight.shape[0]

    key = (bf16, _train_stamp[0], weight._version,
           conv.bias._version if conv.bias is not None else -1,
           bn.weight._version if bn is not None else -1,
           bn.bias._version if bn is not None else -1,
           bn.running_mean._version if bn is not None else -1,
           bn.running_var._version if bn is not None else -1)
    cache = getattr(conv, '_rthd_infer_cache', None)
    if cache is None or cache[0] != key:
        bias_f = (conv.bias.float().contiguous() if conv.bias is not None
                  else torch.zeros(cout, device=x.device,
                                   dtype=torch.float32))
        if bn is not None:
            rstd_run = torch.rsqrt(bn.running_var.float() + bn.eps)
            scale = (bn.weight.float() * rstd_run).contiguous()
            shift = (bn.bias.float()
                     + (bias_f - bn.running_mean.float())
                     * scale).contiguous()
        else:
            scale = torch.ones(cout, device=x.device, dtype=torch.float32)
            shift = bias_f
        if is_stem:
            wpk = _C().pack_weights(_stem_col_weight(weight), False, bf16)
        else:
            wpk = _C().pack_weights(weight, False, bf16)
        conv._rthd_infer_cache = (key, wpk, scale, shift)
        cache = conv._rthd_infer_cache
    _, wpk, scale, shift = cache

    if is_stem:
        assert skc is None
        # unfold + 1x1 MFMA conv (see _ConvBNActFn). MUST go through the
        # dispatcher op — a pybind call here would bake the warmup input's
        # unfolded tensor into the traced export as a constant.
        xcol = ops.stem_im2col(xc, kh, stride, pad)
        return ops.conv_fwd(xcol, wpk, scale, shift, None, 1, 1, 1, 0,
                            cout, act_code)
    return ops.conv_fwd(xc, wpk, scale, shift, skc, kh, kw, stride, pad,
                        cout, act_code)


def conv_bn_act(x, conv, bn, act, act_module=None, training=False,
                skip=None):
    """GPU twin of functional.conv_bn_act (act_module path stays eager)."""
    kh, kw = conv.kernel_size
    stride = conv.stride[0]
    pad = conv.padding[0]
    cin = conv.weight.shape[1]
    # the 3-channel stem uses the direct kernel; everything else (any Cin)
    # goes through the implicit-GEMM kernel
    is_stem = (cin == 3 and conv.weight.shape[0] == 64 and kh == 7
               and kw == 7)
    use_bn = bn is not None
    act_code = ACT_CODE.get(act)
    if act_code is None:
        raise NotImplementedError(f'HIP conv epilogue: activation {act!r}')
    if not training and not torch.is_grad_enabled():
        # fp8 K=128 MFMA path: only when (a) the channel count fills
        # whole 128-k blocks (a Cin=64 layer would burn half its MFMA on
        # zero padding — measured 409 vs 492 TF) and (b) the GEMM is big
        # enough to fill the chip with 128x128 tiles — small-spatial
        # layers run the bf16 path, whose autotune has 64-tile/split-K
        # variants (the fp8 kernel does not); the extra dtype boundary
        # casts there are on tiny tensors
        if (_amp.fp8_enabled() and not is_stem and cin % 128 == 0
                and x.shape[0] * x.shape[2] * x.shape[3] >= 16384):
            y = _conv_infer_fp8(x, conv, bn, act_code, skip)
        else:
            y = _conv_infer(x, conv, bn, act_code, skip, kh, kw, stride,
                            pad, is_stem)
        if act_module is not None:
            y = act_module(y)
        return y
    if training:
        # a training forward mutates BN running stats through raw kernel
        # writes that never bump buffer._version — bump the stamp the
        # inference caches key on, so a later eval forward re-folds
        bump_train_stamp()
    from .functional import bn_momentum
    y = _ConvBNActFn.apply(
        x, conv.weight,
        conv.bias,
        bn.weight if use_bn else None,
        bn.bias if use_bn else None,
        bn.running_mean if use_bn else None,
        bn.running_var if use_bn else None,
        skip,
        kh, kw, stride, pad, act_code, use_bn, training,
        bn_momentum(bn, training) if use_bn else 0.1,
        bn.eps if use_bn else 1e-5,
        is_stem)
    if act_module is not None:
        y = act_module(y)
    return y


# ------------------------------------------------------------ elementwise --

class _AddActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, act_code):
        y = _ops().add_act_fwd(a, b, act_code)
        ctx.act_code = act_code
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dz = _C().add_act_bwd(dy, y, ctx.act_code)
        return dz, dz, None


def add_act(a, b, act='Linear', act_module=None):
    code = ACT_CODE.get(act)
    if code is None:
        raise NotImplementedError(f'HIP add_act: activation {act!r}')
    if not torch.is_grad_enabled():
        y = _ops().add_act_fwd(a, b, code)
    else:
        y = _AddActFn.apply(a, b, code)
    if act_module is not None:
        y = act_module(y)
    return y


# ----------------------------------------------------------------- pools ---

class _MaxPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        need_arg = x.requires_grad
        ctx.hw = (x.shape[2], x.shape[3])
        if not need_arg:
            return _ops().maxpool2x2(x)
        out = _C().pool2x2_fwd(x, True, True)
        ctx.save_for_backward(out[1])
        return out[0]

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        return _C().pool2x2_bwd(dy, arg, True, *ctx.hw)


class _AvgPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.hw = (x.shape[2], x.shape[3])
        return _ops().avgpool2x2(x)

    @staticmethod
    def backward(ctx, dy):
        dummy = torch.empty(0, dtype=torch.uint8, device=dy.device)
        return _C().pool2x2_bwd(dy, dummy, False, *ctx.hw)


class _MaxPoolSameFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k):
        need_arg = x.requires_grad
        ctx.k = k
        if not need_arg:
            return _ops().maxpool_same(x, k)
        out = _C().maxpool_same_fwd(x, k, True)
        ctx.save_for_backward(out[1])
        return out[0]

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        return _C().maxpool_same_bwd(dy, arg, ctx.k), None


def maxpool2x2(x):
    if not torch.is_grad_enabled():
        return _ops().maxpool2x2(x)
    return _MaxPool2x2Fn.apply(x)


def avgpool2x2(x):
    if not torch.is_grad_enabled():
        return _ops().avgpool2x2(x)
    return _AvgPool2x2Fn.apply(x)


def maxpool_same(x, kernel):
    if not torch.is_grad_enabled():
        return _ops().maxpool_same(x, kernel)
    return _MaxPoolSameFn.apply(x, kernel)


# -------------------------------------------------------------- upsample ---

class _Upsample2xAddFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, skip):
        ctx.has_skip = skip is not None
        return _ops().upsample2x_add(x, skip)

    @staticmethod
    def backward(ctx, dy):
        dx = _C().upsample2x_bwd(dy)
        return dx, (dy if ctx.has_skip else None)


def upsample2x_add(x, skip=None):
    if not torch.is_grad_enabled():
        return _ops().upsample2x_add(x, skip)
    if skip is None:
        return _Upsample2xAddFn.apply(x, None)
    return _Upsample2xAddFn.apply(x, skip)


# ------------------------------------------------------------------ loss ---

class _CenterNetLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, phm, poff, psize, ghm, goff, gsize, mask, alpha, beta):
        losses, sums = _C().centernet_loss_fwd(phm, poff, psize, ghm, goff,
                                               gsize, mask, alpha, beta)
        ctx.save_for_backward(phm, poff, psize, ghm, goff, gsize, mask, sums)
        ctx.ab = (alpha, beta)
        return losses[0], losses[1], losses[2]

    @staticmethod
    def backward(ctx, g_hm, g_off, g_size):
        phm, poff, psize, ghm, goff, gsize, mask, sums = ctx.saved_tensors
        alpha, beta = ctx.ab
        gout = torch.stack([g_hm.float(), g_off.float(), g_size.float()])
        dphm, dpoff, dpsize = _C().centernet_loss_bwd(
            phm, poff, psize, ghm, goff, gsize, mask, sums, gout, alpha,
            beta)
        return (dphm, dpoff, dpsize, None, None, None, None, None, None)


def centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                     focal_alpha, focal_beta):
    return _CenterNetLossFn.apply(phm.float(), poff.float(), psize.float(),
                                  ghm, goff, gsize, mask,
                                  float(focal_alpha), float(focal_beta))


class _CenterNetLossLogitsFn(torch.autograd.Function):
    """All-stacks fused form: raw (B,S,C+4,h,w) LOGITS in, [S,3] losses out.

    The heatmap sigmoid (reference train.py:107-111, applied outside the
    network) and the fp32 upcast live inside the kernel; backward produces
    d/d_logit in the logits' dtype. One kernel pair per STEP instead of
    ~10 launches per stack."""

    @staticmethod
    def forward(ctx, out, ghm, goff, gsize, mask, alpha, beta, sig_os):
        out_c = out.contiguous()
        losses, sums = _C().centernet_loss_fused_fwd(
            out_c, ghm, goff, gsize, mask, alpha, beta, sig_os)
        ctx.save_for_backward(out_c, ghm, goff, gsize, mask, sums)
        ctx.meta = (alpha, beta, sig_os)
        return losses

    @staticmethod
    def backward(ctx, glosses):
        out_c, ghm, goff, gsize, mask, sums = ctx.saved_tensors
        alpha, beta, sig_os = ctx.meta
        dout = _C().centernet_loss_fused_bwd(
            out_c, ghm, goff, gsize, mask, sums, glosses.contiguous(),
            alpha, beta, sig_os)
        return dout, None, None, None, None, None, None, None


def centernet_losses_logits(out, ghm, goff, gsize, mask, focal_alpha,
                            focal_beta, sigmoid_offsize=False):
    return _CenterNetLossLogitsFn.apply(out, ghm, goff, gsize, mask,
                                        float(focal_alpha),
                                        float(focal_beta),
                                        bool(sigmoid_offsize))


# ---------------------------------------------------------------- decode ---

def batched_decode(heatmap, offset, wh, scale_factor, topk, pool_size,
                   normalized):
    boxes, clss, scores = _C().decode_fwd(heatmap, offset, wh,
                                          int(scale_factor), int(topk),
                                          int(pool_size), bool(normalized))
    return boxes, clss, scores


_NMS_CAP = 2048  # LDS-resident kernel limit (nms.hip)


def nms_batched(boxes, scores, iou_threshold, conf_th):
    """Batched class-agnostic NMS with the confidence filter folded in:
    one kernel + ONE host sync for the whole batch (the per-image loop
    paid a sync per image). boxes (B,N,4), scores (B,N) ->
    (idx (B,N) int32, counts (B) int32)."""
    if boxes.shape[1] > _NMS_CAP:
        from .eager import nms_batched as eager_nb
        return eager_nb(boxes, scores, iou_threshold, conf_th)
    idx, counts = _C().nms_batched(boxes, scores, float(iou_threshold),
                                   float(conf_th))
    return idx, counts


def nms(boxes, scores, iou_threshold):
    if boxes.shape[0] > _NMS_CAP:
        # configs like --topk 1000 with num_stack>=3 and conf_th=0 exceed
        # the LDS-resident kernel; run the eager O(N^2) suppression instead
        # of aborting (reference used torchvision.ops.nms with no cap)
        from .eager import nms as eager_nms
        return eager_nms(boxes, scores, iou_threshold)
    return _C().nms_fwd(boxes, scores, float(iou_threshold))
