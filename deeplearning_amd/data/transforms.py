"""Image transforms (pure torch + PIL; no torchvision dependency).

Reference parity: the transforms each subproject composes from torchvision
(RandomResizedCrop/flip/normalize — e.g. classification/mnist/dataLoader,
swin dataLoader/build.py) re-implemented on tensors so the hot path
(normalize) can run fused on-GPU.
"""
from __future__ import annotations

import random

import numpy as np
import torch
from PIL import Image

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


def pil_to_tensor(img: Image.Image) -> torch.Tensor:
    arr = np.asarray(img, dtype=np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(arr.copy()).permute(2, 0, 1)
    return t.float().div_(255.0)


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class Resize:
    def __init__(self, size):
        self.size = (size, size) if isinstance(size, int) else tuple(size)

    def __call__(self, img: Image.Image):
        return img.resize(self.size[::-1], Image.BILINEAR)


class CenterCrop:
    def __init__(self, size):
        self.size = (size, size) if isinstance(size, int) else tuple(size)

    def __call__(self, img: Image.Image):
        w, h = img.size
        th, tw = self.size
        x = max((w - tw) // 2, 0)
        y = max((h - th) // 2, 0)
        return img.crop((x, y, x + tw, y + th))


class RandomResizedCrop:
    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3)):
        self.size = (size, size) if isinstance(size, int) else tuple(size)
        self.scale = scale
        self.ratio = ratio

    def __call__(self, img: Image.Image):
        w, h = img.size
        area = w * h
        for _ in range(10):
            target = random.uniform(*self.scale) * area
            ar = random.uniform(*self.ratio)
            cw = int(round((target * ar) ** 0.5))
            ch = int(round((target / ar) ** 0.5))
            if cw <= w and ch <= h:
                x = random.randint(0, w - cw)
                y = random.randint(0, h - ch)
                return img.crop((x, y, x + cw, y + ch)).resize(
                    self.size[::-1], Image.BILINEAR)
        return CenterCrop(self.size)(Resize(min(self.size))(img))


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img: Image.Image):
        if random.random() < self.p:
            return img.transpose(Image.FLIP_LEFT_RIGHT)
        return img


class ToTensor:
    def __call__(self, img):
        return pil_to_tensor(img) if isinstance(img, Image.Image) else img


class Normalize:
    def __init__(self, mean=IMAGENET_MEAN, std=IMAGENET_STD):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, t: torch.Tensor):
        return (t - self.mean) / self.std


def classification_train_transform(img_size=224, rand_augment=False,
                                   ra_magnitude=9.0, ra_mstd=0.5):
    """rand_augment=True inserts RandAugment between the crop and flip
    (swin recipe rand-m9-mstd0.5-inc1; data/autoaugment.py)."""
    ops = [RandomResizedCrop(img_size), RandomHorizontalFlip()]
    if rand_augment:
        from .autoaugment import RandAugment
        ops.append(RandAugment(magnitude=ra_magnitude, mstd=ra_mstd))
    return Compose(ops + [ToTensor(), Normalize()])


def classification_eval_transform(img_size=224, crop_pct=0.875):
    return Compose([Resize(int(img_size / crop_pct)), CenterCrop(img_size),
                    ToTensor(), Normalize()])


class Mixup:
    """Mixup/CutMix for soft-target training (timm Mixup semantics used by
    swin dataLoader/build.py:90-96)."""

    def __init__(self, mixup_alpha=0.8, cutmix_alpha=1.0, prob=1.0,
                 switch_prob=0.5, label_smoothing=0.1, num_classes=1000):
        self.mixup_alpha = mixup_alpha
        self.cutmix_alpha = cutmix_alpha
        self.prob = prob
        self.switch_prob = switch_prob
        self.smoothing = label_smoothing
        self.num_classes = num_classes

    def _one_hot(self, target, lam=1.0, other=None):
        off = self.smoothing / self.num_classes
        on = 1.0 - self.smoothing + off
        y = torch.full((target.shape[0], self.num_classes), off,
                       device=target.device)
        y.scatter_(1, target[:, None], on)
        if other is not None:
            y2 = torch.full_like(y, off)
            y2.scatter_(1, other[:, None], on)
            y = lam * y + (1 - lam) * y2
        return y

    def __call__(self, x, target):
        if random.random() > self.prob:
            return x, self._one_hot(target)
        perm = torch.randperm(x.shape[0], device=x.device)
        use_cutmix = random.random() < self.switch_prob
        if use_cutmix:
            lam = float(np.random.beta(self.cutmix_alpha, self.cutmix_alpha))
            H, W = x.shape[-2:]
            rh, rw = int(H * (1 - lam) ** 0.5), int(W * (1 - lam) ** 0.5)
            cy, cx = random.randint(0, H - 1), random.randint(0, W - 1)
            y1, y2 = max(cy - rh // 2, 0), min(cy + rh // 2, H)
            x1, x2 = max(cx - rw // 2, 0), min(cx + rw // 2, W)
            x[:, :, y1:y2, x1:x2] = x[perm][:, :, y1:y2, x1:x2]
            lam = 1 - (y2 - y1) * (x2 - x1) / (H * W)
        else:
            lam = float(np.random.beta(self.mixup_alpha, self.mixup_alpha))
            x = lam * x + (1 - lam) * x[perm]
        return x, self._one_hot(target, lam, target[perm])


class RandomErasing:
    """Erase a random rectangle with noise (timm RandomErasing semantics used
    by swin dataLoader/build.py; operates on a CHW tensor post-normalize)."""

    def __init__(self, p=0.25, scale=(0.02, 0.33), ratio=(0.3, 3.3)):
        self.p = p
        self.scale = scale
        self.ratio = ratio

    def __call__(self, t: torch.Tensor) -> torch.Tensor:
        if random.random() > self.p:
            return t
        C, H, W = t.shape
        area = H * W
        for _ in range(10):
            target = random.uniform(*self.scale) * area
            ar = random.uniform(*self.ratio)
            eh = int(round((target * ar) ** 0.5))
            ew = int(round((target / ar) ** 0.5))
            if eh < H and ew < W:
                y = random.randint(0, H - eh)
                x = random.randint(0, W - ew)
                t = t.clone()
                t[:, y:y + eh, x:x + ew] = torch.randn(C, eh, ew)
                return t
        return t


class ColorJitter:
    """Brightness/contrast/saturation jitter on a CHW float tensor in [0,1]
    (applied before Normalize)."""

    def __init__(self, brightness=0.4, contrast=0.4, saturation=0.4):
        self.brightness = brightness
        self.contrast = contrast
        self.saturation = saturation

    def __call__(self, t: torch.Tensor) -> torch.Tensor:
        if self.brightness:
            t = t * (1.0 + random.uniform(-self.brightness, self.brightness))
        if self.contrast:
            mean = t.mean()
            f = 1.0 + random.uniform(-self.contrast, self.contrast)
            t = (t - mean) * f + mean
        if self.saturation:
            gray = t.mean(0, keepdim=True)
            f = 1.0 + random.uniform(-self.saturation, self.saturation)
            t = (t - gray) * f + gray
        return t.clamp(0.0, 1.0)
