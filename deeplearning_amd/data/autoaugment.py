"""RandAugment (pure PIL, no timm dependency).

Reference parity: the swin trainer's AUG.AUTO_AUGMENT='rand-m9-mstd0.5-inc1'
(classification/swin_transformer/config.py + timm create_transform call in
dataLoader/build.py) — re-implemented from the published RandAugment paper
(Cubuk et al., 2020) with timm's "inc1" convention: magnitude maps so that
higher m always means a STRONGER augmentation (e.g. Posterize keeps fewer
bits as m grows), and the per-call magnitude is jittered with gaussian std
`mstd` then clamped to [0, 30].
"""
from __future__ import annotations

import random

from PIL import Image, ImageEnhance, ImageOps

_MAX_LEVEL = 30.0
_FILL = (128, 128, 128)


def _affine(img, matrix):
    return img.transform(img.size, Image.AFFINE, matrix,
                         resample=Image.BILINEAR, fillcolor=_FILL)


def _enhance_factor(level):
    # inc1: factor 1.0 +- up to 0.9, symmetric sign chosen at random
    mag = (level / _MAX_LEVEL) * 0.9
    if random.random() > 0.5:
        mag = -mag
    return max(0.1, 1.0 + mag)


def _shear_x(img, level):
    v = (level / _MAX_LEVEL) * 0.3
    if random.random() > 0.5:
        v = -v
    return _affine(img, (1, v, 0, 0, 1, 0))


def _shear_y(img, level):
    v = (level / _MAX_LEVEL) * 0.3
    if random.random() > 0.5:
        v = -v
    return _affine(img, (1, 0, 0, v, 1, 0))


def _translate_x(img, level):
    v = (level / _MAX_LEVEL) * 0.45 * img.size[0]
    if random.random() > 0.5:
        v = -v
    return _affine(img, (1, 0, v, 0, 1, 0))


def _translate_y(img, level):
    v = (level / _MAX_LEVEL) * 0.45 * img.size[1]
    if random.random() > 0.5:
        v = -v
    return _affine(img, (1, 0, 0, 0, 1, v))


def _rotate(img, level):
    deg = (level / _MAX_LEVEL) * 30.0
    if random.random() > 0.5:
        deg = -deg
    return img.rotate(deg, resample=Image.BILINEAR, fillcolor=_FILL)


def _posterize(img, level):
    # inc1: bits 8 -> 4 as level grows (stronger = fewer bits), min 4 like timm
    bits = 8 - int((level / _MAX_LEVEL) * 4)
    return ImageOps.posterize(img, max(4, bits))


def _solarize(img, level):
    thresh = 256 - int((level / _MAX_LEVEL) * 256)
    return ImageOps.solarize(img, max(0, thresh))


def _solarize_add(img, level, thresh=128):
    add = int((level / _MAX_LEVEL) * 110)
    lut = [min(255, i + add) if i < thresh else i for i in range(256)]
    if img.mode == "RGB":
        return img.point(lut * 3)
    return img.point(lut)


OPS = [
    ("AutoContrast", lambda im, lv: ImageOps.autocontrast(im)),
    ("Equalize", lambda im, lv: ImageOps.equalize(im)),
    ("Invert", lambda im, lv: ImageOps.invert(im)),
    ("Rotate", _rotate),
    ("Posterize", _posterize),
    ("Solarize", _solarize),
    ("SolarizeAdd", _solarize_add),
    ("Color", lambda im, lv: ImageEnhance.Color(im).enhance(
        _enhance_factor(lv))),
    ("Contrast", lambda im, lv: ImageEnhance.Contrast(im).enhance(
        _enhance_factor(lv))),
    ("Brightness", lambda im, lv: ImageEnhance.Brightness(im).enhance(
        _enhance_factor(lv))),
    ("Sharpness", lambda im, lv: ImageEnhance.Sharpness(im).enhance(
        _enhance_factor(lv))),
    ("ShearX", _shear_x),
    ("ShearY", _shear_y),
    ("TranslateX", _translate_x),
    ("TranslateY", _translate_y),
]


class RandAugment:
    """Apply `num_ops` randomly chosen ops at magnitude ~N(magnitude, mstd).

    Equivalent policy string: rand-m{magnitude}-mstd{mstd}-inc1 (timm).
    """

    def __init__(self, num_ops: int = 2, magnitude: float = 9.0,
                 mstd: float = 0.5):
        self.num_ops = num_ops
        self.magnitude = magnitude
        self.mstd = mstd

    def __call__(self, img: Image.Image) -> Image.Image:
        for _ in range(self.num_ops):
            name, fn = random.choice(OPS)
            level = self.magnitude
            if self.mstd > 0:
                level = random.gauss(level, self.mstd)
            level = min(_MAX_LEVEL, max(0.0, level))
            img = fn(img, level)
        return img

    def __repr__(self):
        return (f"RandAugment(num_ops={self.num_ops}, "
                f"magnitude={self.magnitude}, mstd={self.mstd})")
