"""Data layer tests: XML parsing, collate contract, synthetic twin,
augmentation box math."""

import os

import numpy as np
import pytest
import torch

from real_time_helmet_detection_amd.data import (VOC, SyntheticVOC,
                                                 TrainAugmentor,
                                                 TestAugmentor, load_dataset,
                                                 parse_voc_xml,
                                                 boxes_from_voc_dict,
                                                 CLASS2INDEX)
from real_time_helmet_detection_amd.data.augment import (fliplr, clip_boxes,
                                                         resize,
                                                         crop_keep_size,
                                                         affine)
from real_time_helmet_detection_amd.config import build_parser

XML = """<annotation>
  <filename>000001.jpg</filename>
  <size><width>200</width><height>100</height><depth>3</depth></size>
  <object><name>hat</name>
    <bndbox><xmin>10</xmin><ymin>20</ymin><xmax>50</xmax><ymax>60</ymax></bndbox>
  </object>
  <object><name>dog</name>
    <bndbox><xmin>60</xmin><ymin>10</ymin><xmax>90</xmax><ymax>40</ymax></bndbox>
  </object>
  <object><name>person</name>
    <bndbox><xmin>100</xmin><ymin>5</ymin><xmax>150</xmax><ymax>95</ymax></bndbox>
  </object>
</annotation>"""


def _make_voc_root(tmp_path, n=3):
    from PIL import Image
    root = tmp_path / 'VOC2028'
    (root / 'JPEGImages').mkdir(parents=True)
    (root / 'Annotations').mkdir()
    (root / 'ImageSets/Main').mkdir(parents=True)
    names = []
    for i in range(n):
        name = '%06d' % i
        names.append(name)
        Image.new('RGB', (200, 100), (i * 20, 100, 50)).save(
            root / 'JPEGImages' / (name + '.jpg'))
        (root / 'Annotations' / (name + '.xml')).write_text(XML)
    (root / 'ImageSets/Main/trainval.txt').write_text('\n'.join(names))
    (root / 'ImageSets/Main/test.txt').write_text('\n'.join(names))
    return str(root)


def test_parse_voc_xml_and_label_quirk():
    import xml.etree.ElementTree as ET
    voc = parse_voc_xml(ET.fromstring(XML))
    boxes, labels = boxes_from_voc_dict(voc)
    assert boxes == [[10, 20, 50, 60], [60, 10, 90, 40], [100, 5, 150, 95]]
    assert labels == [0, 0, 1]  # 'dog' aliases to hat class (SHWD quirk)
    assert voc['annotation']['size']['width'] == '200'
    assert CLASS2INDEX['dog'] == 0


def test_voc_dataset_and_collate(tmp_path):
    root = _make_voc_root(tmp_path)
    ds = VOC(root=root, transform=TestAugmentor(64), image_set='trainval',
             pretrained='imagenet', normalized_coord=False, num_cls=2)
    assert len(ds) == 3
    img, boxes, labels, voc = ds[0]
    assert img.shape == (100, 200, 3)
    assert boxes.shape == (3, 4)

    batch = [ds[i] for i in range(2)]
    out = ds.collate_fn(batch)
    img_t, hm, off, wh, mask, dicts = out
    assert img_t.shape == (2, 3, 64, 64)
    assert hm.shape == (2, 2, 16, 16)
    assert off.shape == (2, 2, 16, 16) and wh.shape == (2, 2, 16, 16)
    assert mask.shape == (2, 1, 16, 16)
    assert mask.sum() > 0
    assert dicts[0]['annotation']['filename'] == '000001.jpg'


def test_load_dataset_factory_synthetic():
    args = build_parser(['--train-flag', '--synthetic', '--imsize', '64',
                         '--multiscale', '32', '64', '16',
                         '--synthetic-size', '5'])
    ds = load_dataset(args)
    assert isinstance(ds, SyntheticVOC)
    assert len(ds) == 5
    out = ds.collate_fn([ds[0], ds[1]])
    assert out[0].shape == (2, 3, 64, 64)


def test_synthetic_deterministic():
    ds = SyntheticVOC(transform=TestAugmentor(64), size=4, imsize=64, seed=1)
    a = ds[2]
    b = ds[2]
    np.testing.assert_array_equal(a[0], b[0])
    np.testing.assert_array_equal(a[1], b[1])


def test_fliplr_box_math():
    img = np.zeros((10, 20, 3), np.uint8)
    boxes = np.array([[2., 1., 6., 5.]])
    rng = np.random.RandomState(0)
    out, b = fliplr(img, boxes, rng, p=1.1)  # always flip
    np.testing.assert_allclose(b, [[14., 1., 18., 5.]])


def test_clip_boxes_removes_and_clips():
    img = np.zeros((10, 10, 3), np.uint8)
    boxes = np.array([[-5., -5., 5., 5.],     # clipped
                      [20., 20., 30., 30.],   # fully outside
                      [2., 2., 8., 8.]])      # untouched
    labels = np.array([0, 1, 1])
    b, l = clip_boxes(img, boxes, labels)
    assert len(b) == 2
    np.testing.assert_allclose(b[0], [0., 0., 5., 5.])
    assert list(l) == [0, 1]


def test_resize_box_scaling():
    img = np.zeros((10, 20, 3), np.uint8)
    boxes = np.array([[2., 1., 6., 5.]])
    out, b = resize(img, boxes, 40)
    assert out.shape == (40, 40, 3)
    np.testing.assert_allclose(b, [[4., 4., 12., 20.]])


def test_affine_identity():
    img = np.random.RandomState(0).randint(0, 255, (16, 16, 3), np.uint8)

    class FixedRng:
        def uniform(self, lo=0.0, hi=1.0):
            return 1.0 if (lo, hi) == (1.0, 1.0) else 0.0
    boxes = np.array([[2., 2., 10., 10.]])
    out, b = affine(img, boxes, FixedRng(), 0.0, (1.0, 1.0))
    np.testing.assert_allclose(b, boxes)


def test_train_augmentor_end_to_end():
    rng = np.random.RandomState(3)
    imgs = [rng.randint(0, 255, (100, 120, 3), np.uint8) for _ in range(2)]
    boxes = [np.array([[10., 10., 50., 60.]]), np.zeros((0, 4))]
    labels = [np.array([1]), np.zeros(0, np.int64)]
    aug = TrainAugmentor(multiscale_flag=True, multiscale=(32, 96, 32),
                         seed=0)
    out_imgs, out_boxes, out_labels = aug(imgs, boxes, labels)
    assert out_imgs[0].shape == out_imgs[1].shape
    assert out_imgs[0].shape[0] in (32, 64)
    for b in out_boxes:
        if len(b):
            assert (b[:, 2] >= b[:, 0]).all() and (b[:, 3] >= b[:, 1]).all()
            assert (b >= 0).all()
            assert (b <= out_imgs[0].shape[0]).all()


def test_train_augmentor_worker_rng_diverges(monkeypatch):
    """Forked DataLoader workers must NOT share one augmentation stream
    (round-1 ADVICE: the parent-process RandomState was inherited
    identically by every worker)."""
    from real_time_helmet_detection_amd.data.augment import TrainAugmentor
    import torch.utils.data as tud

    class _FakeInfo:
        def __init__(self, seed):
            self.seed = seed

    draws = []
    for wseed in (1000, 2000):
        monkeypatch.setattr(tud, 'get_worker_info', lambda s=wseed:
                            _FakeInfo(s))
        aug = TrainAugmentor(seed=7)
        draws.append([aug.rng.uniform() for _ in range(4)])
    assert draws[0] != draws[1]

    # and the same worker seed reproduces the same stream
    monkeypatch.setattr(tud, 'get_worker_info', lambda: _FakeInfo(1000))
    aug = TrainAugmentor(seed=7)
    assert [aug.rng.uniform() for _ in range(4)] == draws[0]


def test_parse_voc_xml_shapes():
    """Rewritten XML parser: single object stays a LIST under
    annotation, repeated tags become lists, empty leaves are dropped."""
    import xml.etree.ElementTree as ET
    from real_time_helmet_detection_amd.data.voc import (parse_voc_xml,
                                                         boxes_from_voc_dict)
    one = ET.fromstring(
        '<annotation><filename>a.jpg</filename><size><width>10</width>'
        '<height>20</height></size><object><name>hat</name><bndbox>'
        '<xmin>1</xmin><ymin>2</ymin><xmax>5</xmax><ymax>6</ymax>'
        '</bndbox></object><empty></empty></annotation>')
    d = parse_voc_xml(one)
    assert isinstance(d['annotation']['object'], list)
    assert d['annotation']['size']['width'] == '10'
    assert 'empty' not in d['annotation']
    boxes, labels = boxes_from_voc_dict(d)
    assert boxes == [[1, 2, 5, 6]] and labels == [0]

    two = ET.fromstring(
        '<annotation><object><name>hat</name><bndbox><xmin>1</xmin>'
        '<ymin>1</ymin><xmax>2</xmax><ymax>2</ymax></bndbox></object>'
        '<object><name>person</name><bndbox><xmin>3</xmin><ymin>3</ymin>'
        '<xmax>4</xmax><ymax>4</ymax></bndbox></object></annotation>')
    boxes, labels = boxes_from_voc_dict(parse_voc_xml(two))
    assert labels == [0, 1] and len(boxes) == 2
