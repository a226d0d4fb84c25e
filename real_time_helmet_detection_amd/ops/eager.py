"""Eager (plain PyTorch) implementations of the framework ops.

These are the CPU execution path and the numerical oracles for the HIP
kernels: every kernel parity test in tests/ compares the gfx950 kernel
against the fp32 eager op here. On a GPU box the hot path must NOT fall back
to these silently — dispatch in ``ops/__init__`` raises if the HIP extension
is missing for a CUDA tensor.
"""

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------- losses ---

def centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                     focal_alpha, focal_beta, eps=1e-7):
    """Return (hm_focal, offset_l1, size_l1) — math per loss.py docstring."""
    neg_inds = 1.0 - mask
    neg_weights = torch.pow(1.0 - ghm, focal_beta)
    pos_loss = torch.log(phm + eps) * torch.pow(1.0 - phm, focal_alpha) * mask
    neg_loss = (torch.log(1.0 - phm + eps) * torch.pow(phm, focal_alpha)
                * neg_weights * neg_inds)
    num_pos = mask.sum().clamp(1, 1e30)
    hm_loss = -(pos_loss.sum(dim=[1, 2, 3]).mean()
                + neg_loss.sum(dim=[1, 2, 3]).mean()) / num_pos

    off_loss = (poff * mask - goff * mask).abs().sum(dim=[1, 2, 3]).mean() / num_pos
    size_loss = (psize * mask - gsize * mask).abs().sum(dim=[1, 2, 3]).mean() / num_pos
    return hm_loss, off_loss, size_loss


# ---------------------------------------------------------------- conv-ish --

def conv2d(x, weight, bias=None, stride=1, padding=0):
    return F.conv2d(x, weight, bias, stride=stride, padding=padding)


def batch_norm(x, running_mean, running_var, weight, bias, training,
               momentum=0.1, eps=1e-5):
    return F.batch_norm(x, running_mean, running_var, weight, bias,
                        training=training, momentum=momentum, eps=eps)


def conv_bn_act(x, weight, bias, bn, act, stride=1, padding=0, training=False):
    """Fused conv -> BN -> activation (the HIP kernel fuses the epilogue)."""
    y = F.conv2d(x, weight, bias, stride=stride, padding=padding)
    if bn is not None:
        y = F.batch_norm(y, bn.running_mean, bn.running_var, bn.weight,
                         bn.bias, training=training, momentum=bn.momentum,
                         eps=bn.eps)
    return apply_act(y, act)


def apply_act(x, act):
    if act is None or act == 'Linear':
        return x
    if act == 'ReLU':
        return F.relu(x)
    if act == 'LeakyReLU':
        return F.leaky_relu(x, 0.01)
    if act == 'PReLU':
        raise ValueError('PReLU carries parameters; use the module path')
    if act == 'Mish':
        return x * torch.tanh(F.softplus(x))
    if act == 'Sigmoid':
        return torch.sigmoid(x)
    if act == 'CELU':
        return F.celu(x)
    raise ValueError(f'unknown activation {act!r}')


def maxpool2x2(x):
    return F.max_pool2d(x, 2, 2)


def avgpool2x2(x):
    return F.avg_pool2d(x, 2, 2)


def maxpool_same(x, kernel):
    """Stride-1 'same' max pool (SPP branches, peak-NMS)."""
    return F.max_pool2d(x, kernel, stride=1, padding=kernel // 2)


def upsample2x_nearest(x):
    return F.interpolate(x, scale_factor=2, mode='nearest')


# ---------------------------------------------------------------- decode ----

def batched_decode(heatmap, offset, wh, scale_factor, topk, pool_size,
                   normalized):
    """Batched peak-mask + top-k decode.

    heatmap: (B,C,h,w) post-sigmoid; offset/wh: (B,2,h,w).
    Returns (boxes (B,topk,4), classes (B,topk), scores (B,topk)) — no
    confidence filtering (callers threshold), so shapes are static for
    tracing and for the HIP kernel.
    """
    b, c, h, w = heatmap.shape
    hw = h * w
    pad = pool_size // 2
    pooled = F.max_pool2d(heatmap, pool_size, stride=1, padding=pad)
    peakmap = heatmap * (pooled == heatmap)

    scores, indices = peakmap.reshape(b, -1).topk(topk, dim=1)
    clss = torch.div(indices, hw, rounding_mode='floor')
    inds = torch.remainder(indices, hw)
    yinds = torch.div(inds, w, rounding_mode='floor')
    xinds = torch.remainder(inds, w)

    flat_off = offset.reshape(b, 2, hw)
    flat_wh = wh.reshape(b, 2, hw)
    xoffs = flat_off[:, 0].gather(1, inds)
    yoffs = flat_off[:, 1].gather(1, inds)
    xsizs = flat_wh[:, 0].gather(1, inds)
    ysizs = flat_wh[:, 1].gather(1, inds)

    if normalized:
        xoffs = xoffs * scale_factor
        yoffs = yoffs * scale_factor
        xsizs = xsizs * w
        ysizs = ysizs * h

    xc = xinds.to(xoffs.dtype) + xoffs
    yc = yinds.to(yoffs.dtype) + yoffs
    sf = float(scale_factor)
    boxes = torch.stack([(xc - xsizs / 2) * sf, (yc - ysizs / 2) * sf,
                         (xc + xsizs / 2) * sf, (yc + ysizs / 2) * sf], dim=2)
    return boxes, clss, scores


# ------------------------------------------------------------------- nms ----

def box_iou(a, b):
    """IoU matrix between (N,4) and (M,4) xyxy boxes."""
    area_a = (a[:, 2] - a[:, 0]).clamp(min=0) * (a[:, 3] - a[:, 1]).clamp(min=0)
    area_b = (b[:, 2] - b[:, 0]).clamp(min=0) * (b[:, 3] - b[:, 1]).clamp(min=0)
    lt = torch.max(a[:, None, :2], b[None, :, :2])
    rb = torch.min(a[:, None, 2:], b[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    inter = wh[..., 0] * wh[..., 1]
    union = area_a[:, None] + area_b[None, :] - inter
    return inter / union.clamp(min=1e-9)


def nms(boxes, scores, iou_threshold):
    """Greedy class-agnostic NMS -> kept indices sorted by score.

    Matches torchvision.ops.nms semantics (the reference eval path,
    evaluate.py:174) without depending on torchvision's compiled op.
    """
    n = boxes.shape[0]
    if n == 0:
        return torch.zeros(0, dtype=torch.long, device=boxes.device)
    order = scores.argsort(descending=True)
    boxes_sorted = boxes[order]
    iou = box_iou(boxes_sorted, boxes_sorted)
    keep_mask = torch.ones(n, dtype=torch.bool, device=boxes.device)
    for i in range(n):
        if keep_mask[i]:
            keep_mask[i + 1:] &= iou[i, i + 1:] <= iou_threshold
    return order[keep_mask]


def nms_batched(boxes, scores, iou_threshold, conf_th):
    """Eager twin of the batched NMS kernel: per image, filter by
    conf_th then greedy NMS; returns (idx (B,N) int32 of KEPT original
    indices, counts (B) int32)."""
    B, N = scores.shape
    idx = torch.zeros(B, N, dtype=torch.int32, device=boxes.device)
    counts = torch.zeros(B, dtype=torch.int32, device=boxes.device)
    for i in range(B):
        keep_conf = scores[i] >= conf_th
        sel = keep_conf.nonzero(as_tuple=False).squeeze(1)
        if sel.numel() == 0:
            continue
        kept = nms(boxes[i][sel], scores[i][sel], iou_threshold)
        orig = sel[kept]
        counts[i] = orig.numel()
        idx[i, :orig.numel()] = orig.to(torch.int32)
    return idx, counts


def soft_nms(boxes, scores, iou_threshold=0.3, sigma=0.5, score_th=0.001):
    """Gaussian soft-NMS (reference evaluate.py:184-243 capability).

    Returns (kept_indices, rescored_scores_for_kept) sorted by decayed score.
    """
    n = boxes.shape[0]
    if n == 0:
        return (torch.zeros(0, dtype=torch.long, device=boxes.device),
                torch.zeros(0, device=boxes.device))
    boxes = boxes.float().clone()
    s = scores.float().clone()
    idx = torch.arange(n, device=boxes.device)
    keep = []
    kept_scores = []
    while idx.numel() > 0:
        top = torch.argmax(s[idx])
        cur = idx[top]
        keep.append(cur.item())
        kept_scores.append(s[cur].item())
        rest = torch.cat([idx[:top], idx[top + 1:]])
        if rest.numel() == 0:
            break
        iou = box_iou(boxes[cur][None], boxes[rest])[0]
        s[rest] = s[rest] * torch.exp(-(iou * iou) / sigma)
        idx = rest[s[rest] > score_th]
    device = boxes.device
    return (torch.tensor(keep, dtype=torch.long, device=device),
            torch.tensor(kept_scores, device=device))
