"""Op dispatch layer.

Every hot op of the detector goes through this module:

- CPU tensors -> ``eager`` (plain PyTorch; also the test oracle).
- CUDA(ROCm) tensors -> ``hip`` (hand-written gfx950 kernels from the in-tree
  extension). A missing extension is a hard error, never a silent fallback.
- ``RTHD_EAGER_GPU=1`` forces eager on GPU for explicit A/B measurement only.

Replaces, MI355X-natively, what the reference delegated to cuDNN/torchvision
(SURVEY.md §2.3): fused conv+BN+act, pooling, upsample, the fused
focal+L1 loss, the maxpool-peak top-k decode and NMS.
"""

import torch

from . import eager
from . import _backend


def _hip(x):
    return x.is_cuda and not _backend.eager_gpu_override()


# ---------------------------------------------------------------- losses ---

def centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                     focal_alpha, focal_beta):
    if _hip(phm):
        from . import hip
        return hip.centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                                    focal_alpha, focal_beta)
    return eager.centernet_losses(phm, poff, psize, ghm, goff, gsize, mask,
                                  focal_alpha, focal_beta)


# ---------------------------------------------------------------- decode ---

def batched_decode(heatmap, offset, wh, scale_factor, topk, pool_size,
                   normalized):
    if _hip(heatmap):
        from . import hip
        return hip.batched_decode(heatmap, offset, wh, scale_factor, topk,
                                  pool_size, normalized)
    return eager.batched_decode(heatmap, offset, wh, scale_factor, topk,
                                pool_size, normalized)


def nms(boxes, scores, iou_threshold):
    if _hip(boxes):
        from . import hip
        return hip.nms(boxes, scores, iou_threshold)
    return eager.nms(boxes, scores, iou_threshold)


def nms_batched(boxes, scores, iou_threshold, conf_th):
    if _hip(boxes):
        from . import hip
        return hip.nms_batched(boxes, scores, iou_threshold, conf_th)
    return eager.nms_batched(boxes, scores, iou_threshold, conf_th)


def soft_nms(boxes, scores, iou_threshold=0.3, sigma=0.5, score_th=0.001):
    # O(N^2) sequential rescoring on <=few hundred boxes: host-side everywhere.
    return eager.soft_nms(boxes, scores, iou_threshold, sigma, score_th)


def available():
    return _backend.ext() is not None
