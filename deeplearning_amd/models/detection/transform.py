"""GeneralizedRCNNTransform: normalize -> resize (min/max side) -> batch with
pad-to-/32, plus postprocess back to original scales.

Reference parity: detection/fasterRcnn/models/transform.py:70-300 (resize
:175, batch_images :175-213, postprocess :214) — re-designed: batching pads
once into a channels-last tensor (NHWC is the native layout for the conv
stack on MI355X).
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F
from torch import nn


def resize_boxes(boxes, original_size, new_size):
    ratios = [torch.tensor(s, dtype=torch.float32, device=boxes.device) /
              torch.tensor(s_orig, dtype=torch.float32, device=boxes.device)
              for s, s_orig in zip(new_size, original_size)]
    ratio_h, ratio_w = ratios
    xmin, ymin, xmax, ymax = boxes.unbind(1)
    return torch.stack((xmin * ratio_w, ymin * ratio_h,
                        xmax * ratio_w, ymax * ratio_h), dim=1)


class ImageList:
    def __init__(self, tensors: torch.Tensor, image_sizes):
        self.tensors = tensors
        self.image_sizes = image_sizes  # [(h, w)] per image, pre-pad

    def to(self, device):
        return ImageList(self.tensors.to(device), self.image_sizes)


class GeneralizedRCNNTransform(nn.Module):
    def __init__(self, min_size=800, max_size=1333,
                 image_mean=(0.485, 0.456, 0.406),
                 image_std=(0.229, 0.224, 0.225), size_divisible=32):
        super().__init__()
        self.min_size = min_size
        self.max_size = max_size
        self.register_buffer("mean",
                             torch.tensor(image_mean).view(1, 3, 1, 1),
                             persistent=False)
        self.register_buffer("std",
                             torch.tensor(image_std).view(1, 3, 1, 1),
                             persistent=False)
        self.size_divisible = size_divisible

    def resize(self, image, target):
        h, w = image.shape[-2:]
        scale = min(self.min_size / min(h, w), self.max_size / max(h, w))
        new_h, new_w = int(round(h * scale)), int(round(w * scale))
        image = F.interpolate(image[None], size=(new_h, new_w),
                              mode="bilinear", align_corners=False)[0]
        if target is not None and "boxes" in target:
            target = dict(target)
            target["boxes"] = resize_boxes(target["boxes"], (h, w),
                                           (new_h, new_w))
        return image, target

    def batch_images(self, images):
        max_h = max(img.shape[-2] for img in images)
        max_w = max(img.shape[-1] for img in images)
        s = self.size_divisible
        max_h = int(math.ceil(max_h / s) * s)
        max_w = int(math.ceil(max_w / s) * s)
        batch = images[0].new_zeros(len(images), 3, max_h, max_w)
        if images[0].is_cuda:
            # NHWC layout for the HIP BN/conv path (the module docstring's
            # stated MI355X layout); CPU stays NCHW-contiguous.
            batch = batch.to(memory_format=torch.channels_last)
        for img, pad in zip(images, batch):
            pad[:, :img.shape[-2], :img.shape[-1]].copy_(img)
        return batch

    def forward(self, images, targets=None):
        images = [img for img in images]
        targets = list(targets) if targets is not None else None
        sizes = []
        for i, img in enumerate(images):
            img = (img - self.mean[0]) / self.std[0]
            img, t = self.resize(img, targets[i] if targets else None)
            images[i] = img
            if targets is not None:
                targets[i] = t
            sizes.append(img.shape[-2:])
        batch = self.batch_images(images)
        return ImageList(batch, [tuple(s) for s in sizes]), targets

    def postprocess(self, detections, image_sizes, original_sizes):
        if self.training:
            return detections
        for det, im_s, o_s in zip(detections, image_sizes, original_sizes):
            det["boxes"] = resize_boxes(det["boxes"], im_s, o_s)
        return detections
