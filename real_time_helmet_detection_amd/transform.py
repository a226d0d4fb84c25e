"""Box <-> heatmap codecs (CenterNet-style).

Behavioral contract from /root/reference/transform.py:4-110:

- ``box2hm``: boxes (xyxy, image pixels) -> per-class center heatmap at
  1/scale_factor resolution + offset map + size map + center mask. The center
  cell is ``int(center/scale)``; the gaussian radius is the center-to-corner
  distance on the feature map (transform.py:42) with sigma = radius/3,
  splatted with elementwise max. ``normalized`` divides offsets by
  scale_factor and sizes by the feature-map dims (transform.py:33-35).
- ``hm2box``: 3x3 (pool_size) max-pool peak mask -> flat top-k over
  (cls, y, x) -> gather offset/size -> boxes in image pixels, confidence
  thresholded.

The encoder here is vectorized numpy (it runs in dataloader workers); the
decoder is pure torch and shape-polymorphic so the same code is traced into
the TorchScript export. The GPU hot path replaces the decoder with the fused
HIP peak+top-k kernel in ``ops`` (same contract, tested against this one).
"""

import numpy as np
import torch


def gaussian_radius(xcen, ycen, xmin, ymin):
    """Center-to-corner distance on the feature map (reference transform.py:42)."""
    return float(((xcen - xmin) ** 2 + (ycen - ymin) ** 2) ** 0.5)


def gaussian2D(shape, sigma=1.0):
    """(2*int(m)+1, 2*int(n)+1) un-normalized gaussian patch."""
    m, n = int(shape[0]), int(shape[1])
    y = np.arange(-m, m + 1, dtype=np.float32)[:, None]
    x = np.arange(-n, n + 1, dtype=np.float32)[None, :]
    return np.exp(-(x * x + y * y) / (2.0 * float(sigma) * float(sigma)))


def draw_gaussian(heatmap, center, radius):
    """Max-splat a gaussian peak of the given (float) radius at center=(x,y)."""
    patch = gaussian2D((radius, radius), sigma=radius / 3.0)
    r = int(radius)
    x, y = int(center[0]), int(center[1])
    height, width = heatmap.shape[:2]
    if x < 0 or y < 0 or x >= width or y >= height:
        return heatmap
    left, right = min(x, r), min(width - x, r + 1)
    top, bottom = min(y, r), min(height - y, r + 1)
    if right + left <= 0 or bottom + top <= 0:
        return heatmap
    view = heatmap[y - top:y + bottom, x - left:x + right]
    gview = patch[r - top:r + bottom, r - left:r + right]
    np.maximum(view, gview, out=view)
    return heatmap


def box2hm(boxes, labels, imsize, scale_factor=4, num_cls=2, normalized=False):
    """Encode boxes into (heatmap, offset, size, mask) numpy maps.

    imsize: (width, height) of the input image. Boxes are xyxy in image
    pixels. Returns float32 arrays shaped (num_cls,h,w), (2,h,w), (2,h,w),
    (1,h,w) with h=height//scale, w=width//scale.
    """
    width, height = imsize[0] // scale_factor, imsize[1] // scale_factor
    heat_map = np.zeros((num_cls, height, width), dtype=np.float32)
    offset_map = np.zeros((2, height, width), dtype=np.float32)
    size_map = np.zeros((2, height, width), dtype=np.float32)
    mask = np.zeros((1, height, width), dtype=np.float32)

    if boxes is None:
        return heat_map, offset_map, size_map, mask

    for box, label in zip(boxes, labels):
        if box is None:
            continue
        xmin, ymin, xmax, ymax = [v / scale_factor for v in box]
        xcen, ycen = (xmax + xmin) / 2.0, (ymax + ymin) / 2.0
        xind, yind = int(xcen), int(ycen)
        if not (0 <= xind < width and 0 <= yind < height):
            continue

        mask[:, yind, xind] = 1.0

        xoff, yoff = xcen - xind, ycen - yind
        xsize, ysize = xmax - xmin, ymax - ymin
        if normalized:
            xoff, yoff = xoff / scale_factor, yoff / scale_factor
            xsize, ysize = xsize / width, ysize / height
        offset_map[0, yind, xind] = xoff
        offset_map[1, yind, xind] = yoff
        size_map[0, yind, xind] = xsize
        size_map[1, yind, xind] = ysize

        radius = gaussian_radius(xcen, ycen, xmin, ymin)
        draw_gaussian(heat_map[int(label)], (xind, yind), radius)

    return heat_map, offset_map, size_map, mask


def peak_mask(heatmap, pool_size=3):
    """Boolean mask of local maxima via pool_size max-pool (stride 1)."""
    pad = pool_size // 2
    pooled = torch.nn.functional.max_pool2d(
        heatmap.unsqueeze(0), pool_size, stride=1, padding=pad).squeeze(0)
    return pooled == heatmap


def hm2box(heatmap, offset, wh, scale_factor=4, topk=10, conf_th=0.3,
           normalized=False, pool_size=3):
    """Decode one sample's maps into (boxes, classes, scores).

    heatmap: (num_cls,h,w) post-sigmoid; offset/wh: (2,h,w).
    Returns boxes (N,4) xyxy image pixels, classes (N,), scores (N,)
    after confidence thresholding. Pure torch -> traceable.
    """
    height, width = heatmap.shape[-2:]
    hw = height * width

    peaks = peak_mask(heatmap, pool_size)
    peakmap = heatmap * peaks

    scores, indices = peakmap.flatten().topk(topk)
    clss = torch.div(indices, hw, rounding_mode='floor')
    inds = torch.remainder(indices, hw)
    yinds = torch.div(inds, width, rounding_mode='floor')
    xinds = torch.remainder(inds, width)

    xoffs = offset[0, yinds, xinds]
    yoffs = offset[1, yinds, xinds]
    xsizs = wh[0, yinds, xinds]
    ysizs = wh[1, yinds, xinds]

    if normalized:
        xoffs = xoffs * scale_factor
        yoffs = yoffs * scale_factor
        xsizs = xsizs * width
        ysizs = ysizs * height

    xc = xinds.to(xoffs.dtype) + xoffs
    yc = yinds.to(yoffs.dtype) + yoffs
    xmin = (xc - xsizs / 2) * scale_factor
    ymin = (yc - ysizs / 2) * scale_factor
    xmax = (xc + xsizs / 2) * scale_factor
    ymax = (yc + ysizs / 2) * scale_factor
    boxes = torch.stack([xmin, ymin, xmax, ymax], dim=1)

    keep = scores >= conf_th
    return boxes[keep], clss[keep], scores[keep]
