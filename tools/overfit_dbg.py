import sys, os
sys.path.insert(0, '/root/repo')
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
torch.manual_seed(0)
ds = SyntheticVOC(transform=TestAugmentor(256), size=4, imsize=256, seed=11)
items = [ds[i] for i in range(4)]
img, hm, off, wh, mask, dicts = ds.collate_fn(items)
img = img.cuda().contiguous(memory_format=CL)
hm, off, wh, mask = (t.cuda() for t in (hm, off, wh, mask))
net = StackedHourglass(1, 64, 6).cuda().to(memory_format=CL)
calc = LossCalculator().cuda()
opt = torch.optim.Adam(net.parameters(), lr=2e-3)
net.train()
for i in range(400):
    opt.zero_grad(set_to_none=True)
    with amp.autocast(True):
        out = net(img)
    total, _ = compute_stack_losses(out, calc, hm, off, wh, mask, 2, False)
    total.backward()
    opt.step()
    if i % 50 == 0:
        print('step', i, 'loss', total.item())
net.eval()
for conf in (0.25, 0.1):
    pred = Prediction(net, topk=20, scale_factor=4, conf_th=conf, nms='nms', nms_th=0.5).cuda()
    with torch.no_grad():
        boxes, clss, scores = pred(img)
    gt, preds = {}, {}
    for i in range(4):
        _, gtb, gtl, voc = items[i]
        name = voc['annotation']['filename']
        gt[name] = (np.asarray(gtb, np.float64), np.asarray(gtl))
        b = boxes[i].cpu().numpy(); c = clss[i].cpu().numpy()[:, None]; s = scores[i].cpu().numpy()[:, None]
        preds[name] = np.hstack([c, s, b]) if len(b) else np.zeros((0, 6))
    res = voc_map(gt, preds)
    print('conf', conf, '->', res)
    print('example scores img0:', scores[0][:6].tolist())
    print('example boxes img0:', boxes[0][:3].tolist(), 'gt:', items[0][1][:3].tolist())
