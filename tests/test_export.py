"""Export path tests: scripted NMS parity, trace-vs-eager equivalence."""

import torch

from real_time_helmet_detection_amd.config import build_parser
from real_time_helmet_detection_amd.models import StackedHourglass
from real_time_helmet_detection_amd.engine.exporter import (Export,
                                                            nms_scripted,
                                                            export_model,
                                                            build_export_module)
from real_time_helmet_detection_amd.ops import eager


def test_nms_scripted_matches_eager():
    torch.manual_seed(0)
    xy = torch.rand(40, 2) * 100
    wh = torch.rand(40, 2) * 30 + 1
    boxes = torch.cat([xy, xy + wh], dim=1)
    scores = torch.rand(40)
    got = nms_scripted(boxes, scores, 0.5)
    want = eager.nms(boxes, scores, 0.5)
    assert sorted(got.tolist()) == sorted(want.tolist())


def test_export_module_forward():
    net = StackedHourglass(num_stack=2, in_ch=16, out_ch=6).eval()
    ex = Export(net, topk=10, scale_factor=4, conf_th=0.0, nms_th=0.5,
                num_cls=2)
    with torch.no_grad():
        boxes, clss, scores = ex(torch.randn(1, 3, 64, 64))
    assert boxes.dim() == 2 and boxes.shape[1] == 4
    assert clss.shape == scores.shape == (boxes.shape[0],)


def test_trace_parity_and_save(tmp_path):
    args = build_parser(['--num-stack', '1', '--hourglass-inch', '16',
                         '--topk', '10'])
    net = StackedHourglass(num_stack=1, in_ch=16, out_ch=6).eval()
    predictor = build_export_module(args, net)
    paths = export_model(predictor, str(tmp_path), imsize=64, do_gpu=False)
    loaded = torch.jit.load(paths['cpu'])
    x = torch.ones(1, 3, 64, 64)
    with torch.no_grad():
        eager_out = predictor(x)
        traced_out = loaded(x)
    for a, b in zip(eager_out, traced_out):
        torch.testing.assert_close(a, b)
