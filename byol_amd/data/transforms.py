"""Two-view SimCLR/BYOL augmentation pipeline, tensor-native.

Replicates the reference's torchvision pipeline
(``/root/reference/main.py:384-397``): RandomResizedCrop(size) ->
HorizontalFlip(p=0.5) -> ColorJitter(0.8s, 0.8s, 0.8s, 0.2s)@p=0.8 ->
Grayscale(p=0.2) -> GaussianBlur(kernel=int(0.1*size), p=0.5); test =
Resize.  And the reference's DALI GPU variant
(``/root/reference/main.py:360-373``): flip p=0.2, saturation 0.2s, no blur.

Implemented directly on float CHW tensors in [0, 1] (no torchvision/PIL
dependency in the hot path): works per-sample on CPU dataloader workers, and
batched on GPU (every op is pure torch and batch-dim agnostic where noted).
Outputs stay in [0, 1] — the engine's lazy sanity pass hard-errors otherwise,
like the reference (``/root/reference/main.py:486-490``).
"""

import math
import random
from typing import List, Optional, Sequence, Tuple

import torch
import torch.nn.functional as F

__all__ = [
    "Compose", "RandomResizedCrop", "RandomHorizontalFlip", "RandomApply",
    "ColorJitter", "RandomGrayscale", "GaussianBlur", "Resize",
    "build_train_and_test_transforms", "TwoViewTransform",
]

_GRAY_W = (0.299, 0.587, 0.114)  # ITU-R 601-2 luma, torchvision's choice


class Compose:
    def __init__(self, transforms: Sequence):
        self.transforms = list(transforms)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        for t in self.transforms:
            x = t(x)
        return x


class RandomResizedCrop:
    def __init__(self, size: Tuple[int, int], scale=(0.08, 1.0),
                 ratio=(3.0 / 4.0, 4.0 / 3.0)):
        self.size = size if isinstance(size, (tuple, list)) else (size, size)
        self.scale = scale
        self.ratio = ratio

    def sample_box(self, height: int, width: int) -> Tuple[int, int, int, int]:
        area = height * width
        log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
        for _ in range(10):
            target_area = area * random.uniform(*self.scale)
            aspect = math.exp(random.uniform(*log_ratio))
            w = int(round(math.sqrt(target_area * aspect)))
            h = int(round(math.sqrt(target_area / aspect)))
            if 0 < w <= width and 0 < h <= height:
                i = random.randint(0, height - h)
                j = random.randint(0, width - w)
                return i, j, h, w
        # fallback: center crop at a valid aspect
        in_ratio = width / height
        if in_ratio < self.ratio[0]:
            w = width
            h = int(round(w / self.ratio[0]))
        elif in_ratio > self.ratio[1]:
            h = height
            w = int(round(h * self.ratio[1]))
        else:
            w, h = width, height
        i = (height - h) // 2
        j = (width - w) // 2
        return i, j, h, w

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        h, w = x.shape[-2:]
        i, j, ch, cw = self.sample_box(h, w)
        crop = x[..., i:i + ch, j:j + cw]
        batched = crop.dim() == 4
        if not batched:
            crop = crop.unsqueeze(0)
        out = F.interpolate(crop, size=self.size, mode="bilinear",
                            align_corners=False, antialias=False)
        out = out.clamp_(0.0, 1.0)  # bilinear can overshoot by fp epsilon
        return out if batched else out.squeeze(0)


class Resize:
    def __init__(self, size: Tuple[int, int]):
        self.size = size if isinstance(size, (tuple, list)) else (size, size)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        batched = x.dim() == 4
        if not batched:
            x = x.unsqueeze(0)
        out = F.interpolate(x, size=self.size, mode="bilinear",
                            align_corners=False, antialias=False)
        out = out.clamp_(0.0, 1.0)  # bilinear can overshoot by fp epsilon
        return out if batched else out.squeeze(0)


class RandomHorizontalFlip:
    def __init__(self, p: float = 0.5):
        self.p = p

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if random.random() < self.p:
            return torch.flip(x, dims=[-1])
        return x


class RandomApply:
    def __init__(self, transforms: Sequence, p: float = 0.5):
        self.transforms = list(transforms)
        self.p = p

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if random.random() < self.p:
            for t in self.transforms:
                x = t(x)
        return x


def _blend(a: torch.Tensor, b, factor: float) -> torch.Tensor:
    return (factor * a + (1.0 - factor) * b).clamp_(0.0, 1.0)


def rgb_to_grayscale(x: torch.Tensor, keep_channels: bool = True):
    r, g, b = x[..., 0:1, :, :], x[..., 1:2, :, :], x[..., 2:3, :, :]
    gray = _GRAY_W[0] * r + _GRAY_W[1] * g + _GRAY_W[2] * b
    if keep_channels:
        gray = gray.expand_as(x) if x.shape[-3] == 3 else gray
    return gray


def adjust_hue(x: torch.Tensor, hue_shift: float) -> torch.Tensor:
    """Shift hue by ``hue_shift`` (in turns, [-0.5, 0.5]) via RGB->HSV->RGB."""
    r, g, b = x[..., 0, :, :], x[..., 1, :, :], x[..., 2, :, :]
    maxc, _ = x.max(dim=-3)
    minc, _ = x.min(dim=-3)
    v = maxc
    deltac = maxc - minc
    s = torch.where(maxc > 0, deltac / maxc.clamp(min=1e-12),
                    torch.zeros_like(maxc))
    dz = deltac.clamp(min=1e-12)
    rc = (maxc - r) / dz
    gc = (maxc - g) / dz
    bc = (maxc - b) / dz
    h = torch.where(r == maxc, bc - gc,
                    torch.where(g == maxc, 2.0 + rc - bc, 4.0 + gc - rc))
    h = (h / 6.0) % 1.0
    h = torch.where(deltac > 0, h, torch.zeros_like(h))
    h = (h + hue_shift) % 1.0
    # HSV -> RGB
    i = torch.floor(h * 6.0)
    f = h * 6.0 - i
    p = v * (1.0 - s)
    q = v * (1.0 - f * s)
    t = v * (1.0 - (1.0 - f) * s)
    i = i.to(torch.long) % 6
    out_r = torch.where(i == 0, v, torch.where(i == 1, q, torch.where(
        i == 2, p, torch.where(i == 3, p, torch.where(i == 4, t, v)))))
    out_g = torch.where(i == 0, t, torch.where(i == 1, v, torch.where(
        i == 2, v, torch.where(i == 3, q, torch.where(i == 4, p, p)))))
    out_b = torch.where(i == 0, p, torch.where(i == 1, p, torch.where(
        i == 2, t, torch.where(i == 3, v, torch.where(i == 4, v, q)))))
    return torch.stack([out_r, out_g, out_b], dim=-3).clamp_(0.0, 1.0)


class ColorJitter:
    """Brightness/contrast/saturation/hue jitter; factors sampled uniformly
    like torchvision's: brightness/contrast/saturation from
    [max(0, 1-s), 1+s], hue from [-h, h]; applied in random order."""

    def __init__(self, brightness=0.0, contrast=0.0, saturation=0.0, hue=0.0):
        self.brightness = brightness
        self.contrast = contrast
        self.saturation = saturation
        self.hue = hue

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        ops = []
        if self.brightness > 0:
            f = random.uniform(max(0.0, 1 - self.brightness),
                               1 + self.brightness)
            ops.append(lambda img, f=f: _blend(img, 0.0, f))
        if self.contrast > 0:
            f = random.uniform(max(0.0, 1 - self.contrast), 1 + self.contrast)
            ops.append(lambda img, f=f: _blend(
                img, rgb_to_grayscale(img).mean(dim=(-1, -2, -3),
                                               keepdim=True), f))
        if self.saturation > 0:
            f = random.uniform(max(0.0, 1 - self.saturation),
                               1 + self.saturation)
            ops.append(lambda img, f=f: _blend(img, rgb_to_grayscale(img), f))
        if self.hue > 0:
            h = random.uniform(-self.hue, self.hue)
            ops.append(lambda img, h=h: adjust_hue(img, h))
        random.shuffle(ops)
        for op in ops:
            x = op(x)
        return x


class RandomGrayscale:
    def __init__(self, p: float = 0.2):
        self.p = p

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if random.random() < self.p:
            return rgb_to_grayscale(x)
        return x


class GaussianBlur:
    """Probabilistic Gaussian blur (the reference's ``datasets.utils
    .GaussianBlur(kernel_size, p)``; sigma ~ U[0.1, 2.0], the SimCLR/BYOL
    recipe). Depthwise conv with a separable kernel."""

    def __init__(self, kernel_size: int, p: float = 0.5,
                 sigma=(0.1, 2.0)):
        k = max(int(kernel_size), 3)
        if k % 2 == 0:
            k += 1
        self.kernel_size = k
        self.p = p
        self.sigma = sigma

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        if random.random() >= self.p:
            return x
        sigma = random.uniform(*self.sigma)
        k = self.kernel_size
        half = (k - 1) / 2.0
        coords = torch.arange(k, dtype=torch.float32, device=x.device) - half
        g = torch.exp(-(coords ** 2) / (2 * sigma * sigma))
        g = g / g.sum()
        c = x.shape[-3]
        batched = x.dim() == 4
        img = x if batched else x.unsqueeze(0)
        kh = g.view(1, 1, k, 1).expand(c, 1, k, 1)
        kw = g.view(1, 1, 1, k).expand(c, 1, 1, k)
        pad = k // 2
        img = F.conv2d(F.pad(img, (0, 0, pad, pad), mode="reflect"), kh,
                       groups=c)
        img = F.conv2d(F.pad(img, (pad, pad, 0, 0), mode="reflect"), kw,
                       groups=c)
        img = img.clamp_(0.0, 1.0)  # kernel sums to 1; clamp fp epsilon
        return img if batched else img.squeeze(0)


class TwoViewTransform:
    """Applies two independent augmentation pipelines to one image —
    the ``(aug1, aug2, label)`` sample contract."""

    def __init__(self, transform):
        self.transform = (transform if isinstance(transform, Compose)
                          else Compose(transform))

    def __call__(self, x: torch.Tensor):
        return self.transform(x), self.transform(x)


def build_train_and_test_transforms(args) -> Tuple[List, List]:
    """Flag-compatible with ``/root/reference/main.py:345-400``. The
    ``dali_*`` task maps to the reference's DALI recipe (flip p=0.2,
    saturation 0.2s, no blur) executed by our GPU pipeline."""
    size = (args.image_size_override, args.image_size_override)
    s = args.color_jitter_strength
    if "dali" in args.task:
        train_transform = [
            RandomResizedCrop(size, scale=(0.08, 1.0),
                              ratio=(3.0 / 4, 4.0 / 3)),
            RandomHorizontalFlip(p=0.2),
            RandomApply([ColorJitter(brightness=0.8 * s, contrast=0.8 * s,
                                     saturation=0.2 * s, hue=0.2 * s)],
                        p=0.8),
            RandomGrayscale(p=0.2),
        ]
    else:
        train_transform = [
            RandomResizedCrop(size),
            RandomHorizontalFlip(p=0.5),
            RandomApply([ColorJitter(brightness=0.8 * s, contrast=0.8 * s,
                                     saturation=0.8 * s, hue=0.2 * s)],
                        p=0.8),
            RandomGrayscale(p=0.2),
            GaussianBlur(kernel_size=int(0.1 * args.image_size_override),
                         p=0.5),
        ]
    test_transform = [Resize(size)]
    return train_transform, test_transform
