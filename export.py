"""TorchScript export CLI (reference /root/reference/export.py:99-152).

python export.py --model-load WEIGHTS/check_point_100.pth [arch flags]

Uses build_parser (NOT get_arguments): arch flags must be passed manually,
matching the reference contract. Produces jit_traced_model_cpu.pth and (when
a GPU is present) jit_traced_model_gpu.pth in --save-path, consumed by the
in-repo C++ LibTorch app (tools/cpp_infer).
"""

import torch

from real_time_helmet_detection_amd.config import build_parser
from real_time_helmet_detection_amd.engine.trainer import load_network
from real_time_helmet_detection_amd.engine.exporter import (
    build_export_module, export_model)

if __name__ == '__main__':
    args = build_parser()
    device = torch.device('cpu')
    network, _, _, _ = load_network(args, device)
    predictor = build_export_module(args, network)
    export_model(predictor, save_dir=args.save_path,
                 imsize=args.imsize or 512)
