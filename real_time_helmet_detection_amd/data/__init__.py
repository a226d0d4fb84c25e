"""Data layer: VOC-XML dataset, synthetic twin, augmentors, loader factory."""

from .voc import (VOC, SyntheticVOC, CLASS2INDEX, INDEX2CLASS, CLASS2COLOR,
                  parse_voc_xml, boxes_from_voc_dict)
from .augment import TrainAugmentor, TestAugmentor


def load_dataset(args):
    """Dataset factory (reference /root/reference/data.py:172-189).

    Train mode builds the TrainAugmentor from the augmentation flags; eval
    mode a square-resize TestAugmentor. ``--synthetic`` (or a missing
    ``--data`` path) selects the in-memory VOC2028-shaped twin.
    """
    if args.train_flag:
        transform = TrainAugmentor(
            crop_percent=tuple(args.crop_percent),
            color_multiply=tuple(args.color_multiply),
            translate_percent=args.translate_percent,
            affine_scale=tuple(args.affine_scale),
            multiscale_flag=args.multiscale_flag,
            multiscale=args.multiscale,
        )
    else:
        transform = TestAugmentor(imsize=args.imsize or args.multiscale[1])

    synthetic = getattr(args, 'synthetic', False) or args.data is None
    if synthetic:
        imsize = args.imsize or args.multiscale[1]
        return SyntheticVOC(
            transform=transform,
            pretrained=args.pretrained,
            normalized_coord=args.normalized_coord,
            num_cls=args.num_cls,
            scale_factor=args.scale_factor,
            size=getattr(args, 'synthetic_size', 512),
            imsize=imsize,
            seed=args.random_seed,
        )
    return VOC(
        root=args.data,
        transform=transform,
        image_set='trainval' if args.train_flag else 'test',
        pretrained=args.pretrained,
        normalized_coord=args.normalized_coord,
        num_cls=args.num_cls,
        scale_factor=args.scale_factor,
    )


__all__ = ['VOC', 'SyntheticVOC', 'TrainAugmentor', 'TestAugmentor',
           'load_dataset', 'CLASS2INDEX', 'INDEX2CLASS', 'CLASS2COLOR',
           'parse_voc_xml', 'boxes_from_voc_dict']
