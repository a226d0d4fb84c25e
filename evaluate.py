"""Single-image demo CLI (reference /root/reference/evaluate.py:245-290).

python evaluate.py --data img.jpg --model-load WEIGHTS/check_point_N.pth \
    --imsize 512 [--conf-th 0.3 --fontsize 10]

Loads the image, predicts, draws colored boxes (+class/score text when
--fontsize > 0), prints the detections and saves ``image.png`` resized back
to the original size under --save-path.
"""

import os
import time

import torch

from real_time_helmet_detection_amd.config import get_arguments
from real_time_helmet_detection_amd.data import INDEX2CLASS
from real_time_helmet_detection_amd.utils import (imload, draw_box,
                                                  write_text, ten2pil)
from real_time_helmet_detection_amd.engine.trainer import load_network
from real_time_helmet_detection_amd.engine.evaluator import Prediction

if __name__ == '__main__':
    args = get_arguments()
    device = torch.device('cpu' if -1 in args.gpu_no else 'cuda')
    imsize = args.imsize or 512

    network, _, _, _ = load_network(args, device)
    predictor = Prediction(
        network=network, topk=args.topk, scale_factor=args.scale_factor,
        conf_th=args.conf_th, nms=args.nms, nms_th=args.nms_th,
        normalized_coord=args.normalized_coord,
        pool_size=args.pool_size).to(device)
    predictor.eval()

    img_ten, origin_size = imload(args.data, imsize, args.pretrained)
    box_ten, cls_ten, score_ten = predictor(img_ten.to(device))
    box_lst = box_ten[0].tolist()
    cls_lst = cls_ten[0].tolist()
    score_lst = score_ten[0].tolist()

    # clamp to the resized image bounds; order the corners so degenerate
    # predictions (x2 < x1 after clamping, possible from an undertrained
    # model) still draw instead of crashing PIL
    def _sane(box):
        x1, y1, x2, y2 = (max(0, min(v, imsize)) for v in box)
        return [min(x1, x2), min(y1, y2), max(x1, x2), max(y1, y2)]
    box_lst = [_sane(box) for box in box_lst]

    img_pil = ten2pil(img_ten[0], args.pretrained)
    for i, (box, cls, score) in enumerate(zip(box_lst, cls_lst, score_lst)):
        img_pil = draw_box(img_pil, box, cls=int(cls))
        if args.fontsize > 0:
            text = '%s: %1.2f' % (INDEX2CLASS[int(cls)], score)
            img_pil = write_text(img_pil, text, (box[0], box[1] - args.fontsize),
                                 fontsize=args.fontsize)
        sx = origin_size[0] / imsize
        sy = origin_size[1] / imsize
        print('%s: Index: %3d, Class: %7s, Score: %1.2f, '
              'Box: %4d, %4d, %4d, %4d'
              % (time.ctime(), i, INDEX2CLASS[int(cls)], score,
                 box[0] * sx, box[1] * sy, box[2] * sx, box[3] * sy))

    img_pil.resize(origin_size).save(
        os.path.join(args.save_path, 'image.png'))
