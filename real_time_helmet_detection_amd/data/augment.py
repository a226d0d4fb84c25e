"""Bounding-box-aware augmentation pipeline (no imgaug dependency).

The reference uses imgaug (dead upstream) with the sequence
Multiply(brightness) -> Affine(translate/scale) -> Crop(keep_size) ->
Fliplr(0.5) -> per-batch multiscale Resize (/root/reference/data.py:127-170).
This module reimplements the same op set on numpy + PIL: each op transforms
the image AND its boxes, out-of-image boxes are removed/clipped after the
geometric ops, and the batch is resized to one (square) target size — random
from range(min,max,step) when multiscale is on, else max.

Boxes are float (N,4) xyxy arrays + int (N,) label arrays throughout.
"""

import numpy as np
from PIL import Image


def _to_pil(img_np):
    return Image.fromarray(img_np.astype(np.uint8))


def multiply_brightness(img_np, rng, lo, hi):
    factor = rng.uniform(lo, hi)
    return np.clip(img_np.astype(np.float32) * factor, 0, 255).astype(np.uint8)


def affine(img_np, boxes, rng, translate_percent, scale_range):
    """Scale about the center + translate; boxes mapped forward."""
    h, w = img_np.shape[:2]
    s = rng.uniform(scale_range[0], scale_range[1])
    tx = rng.uniform(-translate_percent, translate_percent) * w
    ty = rng.uniform(-translate_percent, translate_percent) * h
    cx, cy = w / 2.0, h / 2.0

    # PIL AFFINE maps output(x,y) -> input coords: forward is
    # p' = s*(p - c) + c + t, so in = (out - c - t)/s + c.
    coeffs = (1.0 / s, 0.0, cx - (cx + tx) / s,
              0.0, 1.0 / s, cy - (cy + ty) / s)
    out = _to_pil(img_np).transform((w, h), Image.AFFINE, coeffs,
                                    resample=Image.BILINEAR)
    img_out = np.asarray(out)

    if len(boxes):
        b = boxes.astype(np.float32).copy()
        # forward map: p' = s*(p - c) + c + t
        b[:, [0, 2]] = s * (b[:, [0, 2]] - cx) + cx + tx
        b[:, [1, 3]] = s * (b[:, [1, 3]] - cy) + cy + ty
        boxes = b
    return img_out, boxes


def crop_keep_size(img_np, boxes, rng, lo, hi):
    """Crop a random percent from each side, then resize back (keep_size)."""
    h, w = img_np.shape[:2]
    top = int(rng.uniform(lo, hi) * h)
    bottom = int(rng.uniform(lo, hi) * h)
    left = int(rng.uniform(lo, hi) * w)
    right = int(rng.uniform(lo, hi) * w)
    if top + bottom >= h or left + right >= w:
        return img_np, boxes
    cropped = img_np[top:h - bottom, left:w - right]
    ch, cw = cropped.shape[:2]
    resized = np.asarray(_to_pil(cropped).resize((w, h), Image.BILINEAR))
    if len(boxes):
        b = boxes.astype(np.float32).copy()
        b[:, [0, 2]] = (b[:, [0, 2]] - left) * (w / cw)
        b[:, [1, 3]] = (b[:, [1, 3]] - top) * (h / ch)
        boxes = b
    return resized, boxes


def fliplr(img_np, boxes, rng, p=0.5):
    if rng.uniform() >= p:
        return img_np, boxes
    w = img_np.shape[1]
    out = img_np[:, ::-1].copy()
    if len(boxes):
        b = boxes.astype(np.float32).copy()
        x1 = w - b[:, 2]
        x2 = w - b[:, 0]
        b[:, 0], b[:, 2] = x1, x2
        boxes = b
    return out, boxes


def clip_boxes(img_np, boxes, labels):
    """Remove fully-outside boxes, clip partially-outside ones."""
    if not len(boxes):
        return boxes, labels
    h, w = img_np.shape[:2]
    b = boxes.astype(np.float32)
    keep = (b[:, 2] > 0) & (b[:, 3] > 0) & (b[:, 0] < w) & (b[:, 1] < h)
    b = b[keep]
    labels = np.asarray(labels)[keep]
    b[:, [0, 2]] = b[:, [0, 2]].clip(0, w)
    b[:, [1, 3]] = b[:, [1, 3]].clip(0, h)
    nonempty = (b[:, 2] > b[:, 0]) & (b[:, 3] > b[:, 1])
    return b[nonempty], labels[nonempty]


def resize(img_np, boxes, size):
    """Resize to (size, size) square; boxes scaled per axis."""
    h, w = img_np.shape[:2]
    out = np.asarray(_to_pil(img_np).resize((size, size), Image.BILINEAR))
    if len(boxes):
        b = boxes.astype(np.float32).copy()
        b[:, [0, 2]] *= size / w
        b[:, [1, 3]] *= size / h
        boxes = b
    return out, boxes


class TrainAugmentor:
    """Multiply -> Affine -> Crop(keep_size) -> Fliplr -> clip -> batch resize."""

    def __init__(self, crop_percent=(0.0, 0.1), color_multiply=(1.2, 1.5),
                 translate_percent=0.1, affine_scale=(0.5, 1.5),
                 multiscale_flag=False, multiscale=(320, 512, 64), seed=None):
        self.crop_percent = crop_percent
        self.color_multiply = color_multiply
        self.translate_percent = translate_percent
        self.affine_scale = affine_scale
        self.multiscale_flag = multiscale_flag
        self.multiscale = list(multiscale)
        self._seed = seed
        self._rng = None
        self._worker_seed = None

    @property
    def rng(self):
        """Per-worker RandomState: forked DataLoader workers inherit the
        parent's RNG, so a state created in __init__ would emit IDENTICAL
        augmentation streams in every worker. Derive the state lazily from
        torch's per-worker seed (distinct per worker AND per epoch with
        persistent_workers, via set_epoch -> base_seed reseeding)."""
        info = None
        try:
            import torch.utils.data as tud
            info = tud.get_worker_info()
        except Exception:
            pass
        wseed = info.seed if info is not None else None
        if self._rng is None or wseed != self._worker_seed:
            if wseed is not None:
                seed = (wseed + (self._seed or 0)) % (2 ** 32)
            else:
                seed = self._seed
            self._rng = np.random.RandomState(seed)
            self._worker_seed = wseed
        return self._rng

    def __call__(self, img_lst, boxes_lst, labels_lst):
        rng = self.rng
        out_imgs, out_boxes, out_labels = [], [], []
        for img, boxes, labels in zip(img_lst, boxes_lst, labels_lst):
            boxes = np.asarray(boxes, dtype=np.float32).reshape(-1, 4)
            labels = np.asarray(labels, dtype=np.int64).reshape(-1)
            img = multiply_brightness(img, rng, *self.color_multiply)
            img, boxes = affine(img, boxes, rng, self.translate_percent,
                                self.affine_scale)
            img, boxes = crop_keep_size(img, boxes, rng, *self.crop_percent)
            img, boxes = fliplr(img, boxes, rng)
            boxes, labels = clip_boxes(img, boxes, labels)
            out_imgs.append(img)
            out_boxes.append(boxes)
            out_labels.append(labels)

        if self.multiscale_flag:
            lo, hi, step = self.multiscale
            target = int(rng.choice(np.arange(lo, hi, step)))
        else:
            target = int(self.multiscale[1])

        final_imgs, final_boxes = [], []
        for img, boxes in zip(out_imgs, out_boxes):
            img, boxes = resize(img, boxes, target)
            final_imgs.append(img)
            final_boxes.append(boxes)
        return final_imgs, final_boxes, out_labels


class TestAugmentor:
    """Square resize only (reference data.py:163-170)."""

    def __init__(self, imsize):
        self.imsize = imsize

    def __call__(self, img_lst, boxes_lst, labels_lst):
        out_imgs, out_boxes = [], []
        for img, boxes in zip(img_lst, boxes_lst):
            boxes = np.asarray(boxes, dtype=np.float32).reshape(-1, 4)
            img, boxes = resize(img, boxes, self.imsize)
            out_imgs.append(img)
            out_boxes.append(boxes)
        return out_imgs, out_boxes, [np.asarray(l, dtype=np.int64)
                                     for l in labels_lst]
