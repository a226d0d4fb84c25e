"""Quality-evidence probes through the FULL native pipeline
(HIP bf16 train -> fused decode -> LDS NMS -> in-repo VOC mAP).

The published 88.43 mAP needs the SHWD dataset (no network on the build
machines); these are the available end-to-end proxies:

  overfit  — memorize a handful of images (does the stack optimize the
             detection objective at all): mAP -> 1.0 expected.
  holdout  — train on one synthetic split, report mAP on a DISJOINT-seed
             held-out split every eval interval (does the pipeline
             generalize, not just memorize). VERDICT.md round-1 item 7.

python tools/quality_probe.py [overfit|holdout] [--steps N] ...
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


def load_split(n, size, seed):
    ds = SyntheticVOC(transform=TestAugmentor(size), size=n, imsize=size,
                      seed=seed)
    items = [ds[i] for i in range(n)]
    img, hm, off, wh, mask, dicts = ds.collate_fn(items)
    return items, (img.cuda().contiguous(memory_format=CL),
                   *(t.cuda() for t in (hm, off, wh, mask)))


def eval_map(net, items, img, conf_th=0.15, topk=20, fp8=False):
    net.eval()
    pred = Prediction(net, topk=topk, scale_factor=4, conf_th=conf_th,
                      nms='nms', nms_th=0.5).cuda()
    gt, preds = {}, {}
    import contextlib
    fp8_ctx = amp.fp8_autocast(True) if fp8 else contextlib.nullcontext()
    with torch.no_grad(), amp.autocast(True), fp8_ctx:
        for i0 in range(0, img.shape[0], 32):
            boxes, clss, scores = pred(img[i0:i0 + 32])
            for j in range(len(boxes)):
                i = i0 + j
                _, gtb, gtl, voc = items[i]
                name = voc['annotation']['filename']
                gt[name] = (np.asarray(gtb, np.float64), np.asarray(gtl))
                b = boxes[j].cpu().numpy()
                c = clss[j].cpu().numpy()[:, None]
                s = scores[j].cpu().numpy()[:, None]
                preds[name] = (np.hstack([c, s, b]) if len(b)
                               else np.zeros((0, 6)))
    net.train()
    r = voc_map(gt, preds)
    return r['map'] if r else 0.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('mode', nargs='?', default='overfit',
                    choices=['overfit', 'holdout'])
    ap.add_argument('--steps', type=int, default=2000)
    ap.add_argument('--imgs', type=int, default=8,
                    help='overfit-mode image count')
    ap.add_argument('--train-imgs', type=int, default=256)
    ap.add_argument('--val-imgs', type=int, default=64)
    ap.add_argument('--batch-size', type=int, default=16)
    ap.add_argument('--size', type=int, default=256)
    ap.add_argument('--in-ch', type=int, default=64)
    ap.add_argument('--num-stack', type=int, default=1)
    ap.add_argument('--lr', type=float, default=2e-3)
    ap.add_argument('--eval-every', type=int, default=250)
    args = ap.parse_args()

    torch.manual_seed(0)
    if args.mode == 'overfit':
        tr_items, (img, hm, off, wh, mask) = load_split(
            args.imgs, args.size, seed=11)
        val_items, val_img = tr_items, img
    else:
        # DISJOINT seeds: index streams rng(seed*100003 + i) never collide
        tr_items, (img, hm, off, wh, mask) = load_split(
            args.train_imgs, args.size, seed=1000)
        val_items, (val_img, *_rest) = load_split(
            args.val_imgs, args.size, seed=2000)

    net = StackedHourglass(args.num_stack, args.in_ch, 6).cuda() \
        .to(memory_format=CL)
    calc = LossCalculator().cuda()
    opt = torch.optim.Adam(net.parameters(), lr=args.lr)

    n_train = img.shape[0]
    bs = min(args.batch_size, n_train)
    curve = []
    net.train()
    g = torch.Generator().manual_seed(7)
    for step in range(1, args.steps + 1):
        idx = torch.randperm(n_train, generator=g)[:bs].cuda() \
            if args.mode == 'holdout' else slice(None)
        opt.zero_grad(set_to_none=True)
        with amp.autocast(True):
            out = net(img[idx])
        total, _ = compute_stack_losses(out, calc, hm[idx], off[idx],
                                        wh[idx], mask[idx], 2, False)
        total.backward()
        opt.step()
        if step % args.eval_every == 0:
            m = eval_map(net, val_items, val_img)
            row = {'step': step, 'loss': round(float(total.item()), 4),
                   'val_map@0.5' if args.mode == 'holdout' else 'map@0.5':
                       round(m, 4)}
            if args.mode == 'holdout':
                row['train_map@0.5'] = round(
                    eval_map(net, tr_items[:args.val_imgs],
                             img[:args.val_imgs]), 4)
                if args.in_ch % 128 == 0:
                    # fp8-resident serving quality on the same held-out
                    # split (VERDICT round-1 item 6: fp8 mAP probe)
                    row['val_map@0.5_fp8'] = round(
                        eval_map(net, val_items, val_img, fp8=True), 4)
            curve.append(row)
            print(json.dumps(row), flush=True)
    print(json.dumps({'mode': args.mode, 'final': curve[-1] if curve
                      else None, 'curve': curve}))


if __name__ == '__main__':
    main()
