"""In-repo VOC mAP (replaces the reference's external mAP submodule).

The reference measured quality by writing per-image detection txt files and
running the Cartucho-style mAP tool out of process (README.md:40-44,
evaluate.py:46-54). Here the same metric — per-class average precision at
IoU 0.5 with every-point (area-under-PR-curve) interpolation — is computed
in-repo from the prediction dict and the dataset's ground truth.
"""

from collections import defaultdict

import numpy as np

from ..data.voc import boxes_from_voc_dict, INDEX2CLASS


def _iou_matrix(a, b):
    """IoU between (N,4) and (M,4) xyxy numpy arrays."""
    if len(a) == 0 or len(b) == 0:
        return np.zeros((len(a), len(b)), dtype=np.float64)
    area_a = np.clip(a[:, 2] - a[:, 0], 0, None) * \
        np.clip(a[:, 3] - a[:, 1], 0, None)
    area_b = np.clip(b[:, 2] - b[:, 0], 0, None) * \
        np.clip(b[:, 3] - b[:, 1], 0, None)
    lt = np.maximum(a[:, None, :2], b[None, :, :2])
    rb = np.minimum(a[:, None, 2:], b[None, :, 2:])
    wh = np.clip(rb - lt, 0, None)
    inter = wh[..., 0] * wh[..., 1]
    union = area_a[:, None] + area_b[None, :] - inter
    return inter / np.clip(union, 1e-9, None)


def average_precision(recall, precision):
    """Every-point interpolated AP (VOC2010+/Cartucho method)."""
    mrec = np.concatenate([[0.0], recall, [1.0]])
    mpre = np.concatenate([[0.0], precision, [0.0]])
    for i in range(len(mpre) - 2, -1, -1):
        mpre[i] = max(mpre[i], mpre[i + 1])
    idx = np.where(mrec[1:] != mrec[:-1])[0]
    return float(np.sum((mrec[idx + 1] - mrec[idx]) * mpre[idx + 1]))


def voc_map(ground_truth, predictions, iou_threshold=0.5,
            class_names=None):
    """Compute per-class AP + mAP.

    ground_truth: {filename: (boxes (N,4) ndarray, labels (N,) ndarray)}
    predictions:  {filename: (M,6) ndarray [cls, score, x1, y1, x2, y2]}
    Returns {'ap': {class_name: AP}, 'map': mAP}.
    """
    class_names = class_names or INDEX2CLASS
    classes = sorted({int(l) for _, labels in ground_truth.values()
                      for l in labels})
    if not classes:
        return None

    aps = {}
    for cls in classes:
        gt_by_img = {}
        n_gt = 0
        for fname, (boxes, labels) in ground_truth.items():
            sel = np.asarray(labels) == cls
            gtb = np.asarray(boxes, dtype=np.float64).reshape(-1, 4)[sel]
            gt_by_img[fname] = {'boxes': gtb,
                                'used': np.zeros(len(gtb), dtype=bool)}
            n_gt += len(gtb)

        dets = []  # (score, fname, box)
        for fname, pred in predictions.items():
            if fname not in gt_by_img or len(pred) == 0:
                continue
            pred = np.asarray(pred)
            sel = pred[:, 0].astype(int) == cls
            for row in pred[sel]:
                dets.append((float(row[1]), fname, row[2:6]))
        dets.sort(key=lambda t: -t[0])

        tp = np.zeros(len(dets))
        fp = np.zeros(len(dets))
        for i, (score, fname, box) in enumerate(dets):
            gt = gt_by_img[fname]
            if len(gt['boxes']) == 0:
                fp[i] = 1
                continue
            ious = _iou_matrix(box[None, :], gt['boxes'])[0]
            j = int(np.argmax(ious))
            if ious[j] >= iou_threshold and not gt['used'][j]:
                tp[i] = 1
                gt['used'][j] = True
            else:
                fp[i] = 1

        if n_gt == 0:
            continue
        ctp = np.cumsum(tp)
        cfp = np.cumsum(fp)
        recall = ctp / n_gt
        precision = ctp / np.clip(ctp + cfp, 1e-9, None)
        name = class_names.get(cls, str(cls)) if hasattr(class_names, 'get') \
            else str(cls)
        aps[name] = average_precision(recall, precision)

    if not aps:
        return None
    return {'ap': aps, 'map': float(np.mean(list(aps.values())))}


def voc_map_from_dataset(dataset, predictions, iou_threshold=0.5):
    """Build the GT dict from a VOC-style dataset's voc_dicts and score."""
    ground_truth = {}
    for i in range(len(dataset)):
        _, boxes, labels, voc_dict = dataset[i]
        fname = voc_dict['annotation']['filename']
        ground_truth[fname] = (np.asarray(boxes, dtype=np.float64),
                               np.asarray(labels))
    return voc_map(ground_truth, predictions, iou_threshold)
