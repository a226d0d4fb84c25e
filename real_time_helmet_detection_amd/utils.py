"""Small utilities: pickle I/O, meters, normalizers, visualization.

Capability parity with /root/reference/utils.py:9-94 (save/load_pickle,
AverageMeter, ten2pil, draw_box, write_text, get_normalizer, blend_heatmap,
imload) — re-implemented from the behavioral contract, not translated.
"""

import pickle

import numpy as np
import torch

try:
    from PIL import Image, ImageDraw, ImageFont
    _HAS_PIL = True
except Exception:  # pragma: no cover - PIL is present in the target image
    _HAS_PIL = False

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)
SCRATCH_MEAN = (0.5, 0.5, 0.5)
SCRATCH_STD = (0.5, 0.5, 0.5)


def save_pickle(path, obj):
    with open(path, 'wb') as f:
        pickle.dump(obj, f)


def load_pickle(path):
    with open(path, 'rb') as f:
        return pickle.load(f)


class AverageMeter:
    """Running average of a scalar (reference utils.py:19-31)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.val = 0.0
        self.sum = 0.0
        self.count = 0
        self.avg = 0.0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / max(self.count, 1)


class Normalizer:
    """Channel-wise normalize / denormalize as a callable on CHW tensors."""

    def __init__(self, mean, std, inverse=False):
        self.mean = torch.tensor(mean, dtype=torch.float32).view(-1, 1, 1)
        self.std = torch.tensor(std, dtype=torch.float32).view(-1, 1, 1)
        self.inverse = inverse

    def __call__(self, x):
        mean = self.mean.to(device=x.device, dtype=x.dtype)
        std = self.std.to(device=x.device, dtype=x.dtype)
        if self.inverse:
            return x * std + mean
        return (x - mean) / std


def get_normalizer(pretrained='imagenet', inverse=False):
    """ImageNet vs scratch normalization (reference utils.py:55-68)."""
    if pretrained == 'imagenet':
        return Normalizer(IMAGENET_MEAN, IMAGENET_STD, inverse)
    return Normalizer(SCRATCH_MEAN, SCRATCH_STD, inverse)


def ten2pil(tensor, pretrained='imagenet'):
    """Batch tensor -> single PIL image grid with denormalization."""
    denorm = get_normalizer(pretrained, inverse=True)
    if tensor.dim() == 4:
        imgs = [denorm(t.detach().float().cpu()) for t in tensor]
        n = len(imgs)
        h, w = imgs[0].shape[-2:]
        grid = torch.zeros(3, h, w * n)
        for i, im in enumerate(imgs):
            grid[:, :, i * w:(i + 1) * w] = im[:3]
    else:
        grid = denorm(tensor.detach().float().cpu())
    arr = (grid.clamp(0, 1) * 255).byte().permute(1, 2, 0).numpy()
    if not _HAS_PIL:
        return arr
    return Image.fromarray(arr)


# distinct box colors per class
_CLS_COLORS = [(255, 64, 64), (64, 160, 255), (64, 255, 96), (255, 224, 64),
               (224, 64, 255), (64, 255, 255)]


def draw_box(image, box, cls=0, width=2):
    """Draw one xyxy box on a PIL image in the class color. Corners are
    ordered first — degenerate predictions (x2 < x1) must not crash the
    renderer."""
    draw = ImageDraw.Draw(image)
    color = _CLS_COLORS[int(cls) % len(_CLS_COLORS)]
    x1, y1, x2, y2 = map(float, box)
    draw.rectangle([min(x1, x2), min(y1, y2), max(x1, x2), max(y1, y2)],
                   outline=color, width=width)
    return image


def write_text(image, text, pos, fontsize=10):
    """Write text at pos (used by the demo to annotate class/score)."""
    draw = ImageDraw.Draw(image)
    try:
        font = ImageFont.load_default()
    except Exception:  # pragma: no cover
        font = None
    draw.text(tuple(map(float, pos)), text, fill=(255, 255, 255), font=font)
    return image


def blend_heatmap(image_tensor, heatmap, pretrained='imagenet', alpha=0.3):
    """Overlay per-class heatmaps on the (denormalized) image.

    image_tensor: (3,H,W) normalized; heatmap: (C,h,w) in [0,1].
    Returns a PIL image (or ndarray without PIL). Reference utils.py:70-85.
    """
    denorm = get_normalizer(pretrained, inverse=True)
    img = denorm(image_tensor.detach().float().cpu()).clamp(0, 1)
    hm = heatmap.detach().float().cpu().clamp(0, 1)
    c, h, w = hm.shape
    hm_up = torch.nn.functional.interpolate(
        hm[None], size=img.shape[-2:], mode='nearest')[0]
    overlay = img.clone()
    for ci in range(c):
        color = torch.tensor(_CLS_COLORS[ci % len(_CLS_COLORS)],
                             dtype=torch.float32) / 255.0
        overlay = overlay * (1 - alpha * hm_up[ci]) + \
            color.view(3, 1, 1) * (alpha * hm_up[ci])
    arr = (overlay.clamp(0, 1) * 255).byte().permute(1, 2, 0).numpy()
    if not _HAS_PIL:
        return arr
    return Image.fromarray(arr)


def imload(path, imsize=512, pretrained='imagenet'):
    """Load an image file -> (1,3,imsize,imsize) normalized tensor + orig size.

    Reference utils.py:87-94 returns the resized normalized tensor; we also
    return the original (w, h) so the demo can resize back.
    """
    img = Image.open(path).convert('RGB')
    ow, oh = img.size
    img = img.resize((imsize, imsize), Image.BILINEAR)
    arr = torch.from_numpy(np.asarray(img).copy()).float().permute(2, 0, 1) / 255.0
    norm = get_normalizer(pretrained)
    return norm(arr).unsqueeze(0), (ow, oh)
