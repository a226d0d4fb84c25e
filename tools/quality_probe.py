"""Quality-evidence probe: overfit the full pipeline (HIP train -> decode
-> NMS -> VOC mAP) on synthetic VOC2028-shaped images and report the mAP
curve. The published 88.43 mAP needs the SHWD dataset (no network on the
build machines), so this is the available proxy that the training stack
optimizes the detection objective end to end.

python tools/quality_probe.py [--steps 2000] [--imgs 8] [--size 256]
"""
import argparse
import json
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from real_time_helmet_detection_amd.models import StackedHourglass
from real_time_helmet_detection_amd.loss import LossCalculator
from real_time_helmet_detection_amd.engine.trainer import compute_stack_losses
from real_time_helmet_detection_amd.engine.evaluator import Prediction
from real_time_helmet_detection_amd.engine.metrics import voc_map
from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
from real_time_helmet_detection_amd import amp

CL = torch.channels_last


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--steps', type=int, default=2000)
    ap.add_argument('--imgs', type=int, default=8)
    ap.add_argument('--size', type=int, default=256)
    ap.add_argument('--eval-every', type=int, default=250)
    args = ap.parse_args()

    torch.manual_seed(0)
    ds = SyntheticVOC(transform=TestAugmentor(args.size), size=args.imgs,
                      imsize=args.size, seed=11)
    items = [ds[i] for i in range(args.imgs)]
    img, hm, off, wh, mask, dicts = ds.collate_fn(items)
    img = img.cuda().contiguous(memory_format=CL)
    hm, off, wh, mask = (t.cuda() for t in (hm, off, wh, mask))

    net = StackedHourglass(1, 64, 6).cuda().to(memory_format=CL)
    calc = LossCalculator().cuda()
    opt = torch.optim.Adam(net.parameters(), lr=2e-3)

    def eval_map():
        net.eval()
        pred = Prediction(net, topk=20, scale_factor=4, conf_th=0.15,
                          nms='nms', nms_th=0.5).cuda()
        with torch.no_grad():
            boxes, clss, scores = pred(img)
        gt, preds = {}, {}
        for i in range(args.imgs):
            _, gtb, gtl, voc = items[i]
            name = voc['annotation']['filename']
            gt[name] = (np.asarray(gtb, np.float64), np.asarray(gtl))
            b = boxes[i].cpu().numpy()
            c = clss[i].cpu().numpy()[:, None]
            s = scores[i].cpu().numpy()[:, None]
            preds[name] = (np.hstack([c, s, b]) if len(b)
                           else np.zeros((0, 6)))
        net.train()
        r = voc_map(gt, preds)
        return r['map'] if r else 0.0

    curve = []
    net.train()
    for step in range(1, args.steps + 1):
        opt.zero_grad(set_to_none=True)
        with amp.autocast(True):
            out = net(img)
        total, _ = compute_stack_losses(out, calc, hm, off, wh, mask, 2,
                                        False)
        total.backward()
        opt.step()
        if step % args.eval_every == 0:
            m = eval_map()
            curve.append({'step': step, 'loss': float(total.item()),
                          'map@0.5': round(m, 4)})
            print(json.dumps(curve[-1]))
    print(json.dumps({'final': curve[-1] if curve else None,
                      'curve': curve}))


if __name__ == '__main__':
    main()
