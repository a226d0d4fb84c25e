"""VOC-XML detection dataset (SHWD / VOC2028 layout) + synthetic twin.

Capability parity with /root/reference/data.py:22-125:

- Reads ``JPEGImages/``, ``Annotations/*.xml`` and the split file
  ``ImageSets/Main/{trainval,test}.txt``.
- Label map keeps the SHWD quirk: {'hat': 0, 'person': 1, 'dog': 0} — 'dog'
  aliases to the hat class (reference data.py:17).
- ``__getitem__`` returns raw (img_np, boxes, labels, voc_dict); ALL
  augmentation + heatmap encoding happen batch-wise in ``collate_fn`` so the
  heavy CPU work runs inside DataLoader workers.
- ``collate_fn`` -> (img, heatmap, offset, wh, mask, voc_dict_list); the voc
  dicts carry filename + original size for eval-time box rescaling.

``SyntheticVOC`` generates VOC2028-shaped data in memory (random images,
random plausible hat/person boxes, same voc_dict contract) for the
no-network benchmark configs (BASELINE.json) and CPU tests.
"""

import os
import time
import xml.etree.ElementTree as ET

import numpy as np
import torch

from ..transform import box2hm
from ..utils import get_normalizer

CLASS2INDEX = {'hat': 0, 'person': 1, 'dog': 0}
INDEX2CLASS = {0: 'hat', 1: 'person'}
CLASS2COLOR = {0: (255, 0, 0), 1: (0, 255, 0)}


def _element_value(node):
    """XML element -> python value: leaf tags give their stripped text,
    interior tags give a dict of child values. A tag repeated among its
    siblings collapses into a list; under <annotation>, 'object' stays a
    list even when a single <object> is present (the eval driver and
    boxes_from_voc_dict rely on that — reference data.py:65-80 contract)."""
    children = list(node)
    if not children:
        return (node.text or '').strip()
    grouped = {}
    for child in children:
        v = _element_value(child)
        if v == '':  # empty leaf tags carry no information — drop them
            continue
        grouped.setdefault(child.tag, []).append(v)
    out = {}
    for tag, values in grouped.items():
        keep_list = len(values) > 1 or (node.tag == 'annotation'
                                        and tag == 'object')
        out[tag] = values if keep_list else values[0]
    return out


def parse_voc_xml(node):
    """XML root -> {tag: nested dict} (same shape the reference produces)."""
    value = _element_value(node)
    return {node.tag: value} if value != '' else {}


def boxes_from_voc_dict(voc_dict):
    """Extract (boxes list, label-index list) from a parsed voc dict."""
    box_lst, id_lst = [], []
    objs = voc_dict['annotation']['object']
    if isinstance(objs, dict):
        objs = [objs]
    for obj in objs:
        id_lst.append(CLASS2INDEX[obj['name'].lower()])
        bb = obj['bndbox']
        box_lst.append([int(float(bb['xmin'])), int(float(bb['ymin'])),
                        int(float(bb['xmax'])), int(float(bb['ymax']))])
    return box_lst, id_lst


class _CollateMixin:
    """Shared collate: batch augmentation -> heatmap encode -> tensors."""

    def collate_fn(self, batch):
        imgs, boxes_lst, labels_lst, voc_dicts = zip(*batch)
        imgs, boxes_lst, labels_lst = self.transform(
            list(imgs), list(boxes_lst), list(labels_lst))

        h, w = imgs[0].shape[:2]
        hm_l, off_l, wh_l, mask_l = [], [], [], []
        for boxes, labels in zip(boxes_lst, labels_lst):
            hm, off, wh, mask = box2hm(
                boxes if len(boxes) else None, labels, (w, h),
                scale_factor=self.scale_factor, num_cls=self.num_cls,
                normalized=self.normalized_coord)
            hm_l.append(hm)
            off_l.append(off)
            wh_l.append(wh)
            mask_l.append(mask)

        img_t = torch.stack([
            self.normalize(torch.from_numpy(
                np.ascontiguousarray(im.transpose(2, 0, 1))).float() / 255.0)
            for im in imgs])
        return (img_t,
                torch.from_numpy(np.stack(hm_l)),
                torch.from_numpy(np.stack(off_l)),
                torch.from_numpy(np.stack(wh_l)),
                torch.from_numpy(np.stack(mask_l)),
                list(voc_dicts))


class VOC(_CollateMixin, torch.utils.data.Dataset):
    def __init__(self, root, transform, image_set, pretrained,
                 normalized_coord, num_cls, scale_factor=4):
        from PIL import Image  # local import: workers re-import lazily
        self._Image = Image
        self.transform = transform
        self.image_set = image_set
        self.normalize = get_normalizer(pretrained=pretrained)
        self.normalized_coord = normalized_coord
        self.num_cls = num_cls
        self.scale_factor = scale_factor

        image_dir = os.path.join(root, 'JPEGImages')
        annotation_dir = os.path.join(root, 'Annotations')
        split_f = os.path.join(root, 'ImageSets/Main',
                               image_set.rstrip('\n') + '.txt')
        with open(split_f, 'r') as f:
            names = [x.strip() for x in f.readlines() if x.strip()]
        self.images = [os.path.join(image_dir, x + '.jpg') for x in names]
        self.annotations = [os.path.join(annotation_dir, x + '.xml')
                            for x in names]
        assert len(self.images) == len(self.annotations)
        print('%s: %d images are loaded from %s'
              % (time.ctime(), len(self.images), root))

    def __len__(self):
        return len(self.images)

    def __getitem__(self, index):
        img = np.asarray(
            self._Image.open(self.images[index]).convert('RGB'))
        voc_dict = parse_voc_xml(ET.parse(self.annotations[index]).getroot())
        boxes, labels = boxes_from_voc_dict(voc_dict)
        return img, np.asarray(boxes, dtype=np.float32).reshape(-1, 4), \
            np.asarray(labels, dtype=np.int64), voc_dict


class SyntheticVOC(_CollateMixin, torch.utils.data.Dataset):
    """VOC2028-shaped synthetic data, deterministic per (seed, index).

    Images are colored-noise backgrounds with brighter rectangles where the
    'objects' are (so heatmap targets correlate with pixels); per image
    1..6 boxes with plausible hat/person aspect ratios.
    """

    def __init__(self, transform, pretrained='imagenet',
                 normalized_coord=False, num_cls=2, scale_factor=4,
                 size=512, imsize=512, seed=777):
        self.transform = transform
        self.normalize = get_normalizer(pretrained=pretrained)
        self.normalized_coord = normalized_coord
        self.num_cls = num_cls
        self.scale_factor = scale_factor
        self.size = size
        self.imsize = imsize
        self.seed = seed

    def __len__(self):
        return self.size

    def __getitem__(self, index):
        rng = np.random.RandomState((self.seed * 100003 + index)
                                    % (2 ** 32))
        h = w = self.imsize
        img = rng.randint(0, 128, size=(h, w, 3), dtype=np.uint8)
        n = rng.randint(1, 7)
        boxes, labels = [], []
        for _ in range(n):
            cls = int(rng.randint(0, self.num_cls))
            # hat-ish: small square-ish; person-ish: tall
            if cls == 0:
                bw = rng.randint(w // 16, w // 5)
                bh = int(bw * rng.uniform(0.8, 1.2))
            else:
                bh = rng.randint(h // 8, h // 2)
                bw = int(bh * rng.uniform(0.3, 0.6))
            x1 = rng.randint(0, max(1, w - bw))
            y1 = rng.randint(0, max(1, h - bh))
            x2, y2 = min(x1 + bw, w - 1), min(y1 + bh, h - 1)
            if x2 <= x1 + 2 or y2 <= y1 + 2:
                continue
            img[y1:y2, x1:x2] = np.minimum(
                img[y1:y2, x1:x2].astype(np.int32) + 96 + 32 * cls, 255
            ).astype(np.uint8)
            boxes.append([x1, y1, x2, y2])
            labels.append(cls)
        voc_dict = {'annotation': {
            'filename': 'synthetic_%06d.jpg' % index,
            'size': {'width': str(w), 'height': str(h), 'depth': '3'},
            'object': [{'name': INDEX2CLASS[l],
                        'bndbox': {'xmin': str(b[0]), 'ymin': str(b[1]),
                                   'xmax': str(b[2]), 'ymax': str(b[3])}}
                       for b, l in zip(boxes, labels)],
        }}
        return img, np.asarray(boxes, dtype=np.float32).reshape(-1, 4), \
            np.asarray(labels, dtype=np.int64), voc_dict
