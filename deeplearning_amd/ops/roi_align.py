"""RoIAlign (torchvision semantics) + MultiScaleRoIAlign (FPN level mapper).

Reference parity: torchvision.ops.MultiScaleRoIAlign in fasterRcnn
(models/faster_rcnn.py:8,305-309). GPU path: csrc/roialign.hip; CPU reference
is a pure-PyTorch bilinear implementation used by the parity tests.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import ext, use_hip


def _roi_align_eager(input, rois, output_size, spatial_scale, sampling_ratio, aligned):
    """Pure-PyTorch RoIAlign (exact semantics, slow; CPU reference)."""
    PH, PW = output_size
    R = rois.shape[0]
    N, C, H, W = input.shape
    out = input.new_zeros((R, C, PH, PW))
    off = 0.5 if aligned else 0.0
    for r in range(R):
        b = int(rois[r, 0].item())
        x1 = rois[r, 1].item() * spatial_scale - off
        y1 = rois[r, 2].item() * spatial_scale - off
        x2 = rois[r, 3].item() * spatial_scale - off
        y2 = rois[r, 4].item() * spatial_scale - off
        rw, rh = x2 - x1, y2 - y1
        if not aligned:
            rw, rh = max(rw, 1.0), max(rh, 1.0)
        bh, bw = rh / PH, rw / PW
        gh = sampling_ratio if sampling_ratio > 0 else max(1, int(torch.tensor(rh / PH).ceil()))
        gw = sampling_ratio if sampling_ratio > 0 else max(1, int(torch.tensor(rw / PW).ceil()))
        for ph in range(PH):
            for pw in range(PW):
                acc = input.new_zeros(C)
                cnt = 0
                for iy in range(gh):
                    y = y1 + ph * bh + (iy + 0.5) * bh / gh
                    for ix in range(gw):
                        x = x1 + pw * bw + (ix + 0.5) * bw / gw
                        cnt += 1
                        if y < -1.0 or y > H or x < -1.0 or x > W:
                            continue
                        yy, xx = max(y, 0.0), max(x, 0.0)
                        y0, x0 = int(yy), int(xx)
                        y1i, x1i = y0 + 1, x0 + 1
                        if y0 >= H - 1:
                            y0 = y1i = H - 1
                            yy = float(y0)
                        if x0 >= W - 1:
                            x0 = x1i = W - 1
                            xx = float(x0)
                        ly, lx = yy - y0, xx - x0
                        hy, hx = 1 - ly, 1 - lx
                        acc += (hy * hx * input[b, :, y0, x0] + hy * lx * input[b, :, y0, x1i]
                                + ly * hx * input[b, :, y1i, x0] + ly * lx * input[b, :, y1i, x1i])
                out[r, :, ph, pw] = acc / max(cnt, 1)
    return out


class _RoIAlignFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, rois, output_size, spatial_scale, sampling_ratio, aligned):
        input = input.contiguous()
        ctx.save_for_backward(rois)
        ctx.params = (input.shape, spatial_scale, sampling_ratio, aligned)
        return ext().roialign_fwd(input, rois, output_size[0], output_size[1],
                                  spatial_scale, sampling_ratio, aligned)

    @staticmethod
    def backward(ctx, grad_out):
        (rois,) = ctx.saved_tensors
        (N, C, H, W), ss, sr, al = ctx.params
        grad_in = ext().roialign_bwd(grad_out, rois, N, C, H, W, ss, sr, al)
        return grad_in, None, None, None, None, None


def roi_align(input: torch.Tensor, rois: torch.Tensor, output_size,
              spatial_scale: float = 1.0, sampling_ratio: int = -1,
              aligned: bool = False) -> torch.Tensor:
    """rois: (R,5) [batch_idx, x1, y1, x2, y2] in input-image coordinates."""
    if isinstance(output_size, int):
        output_size = (output_size, output_size)
    if use_hip(input):
        return _RoIAlignFn.apply(input, rois.to(torch.float32), output_size,
                                 spatial_scale, sampling_ratio, aligned)
    return _roi_align_eager(input, rois, output_size, spatial_scale,
                            sampling_ratio, aligned)


class MultiScaleRoIAlign(nn.Module):
    """FPN-level-aware RoIAlign (torchvision semantics: canonical level mapping
    k = floor(4 + log2(sqrt(area)/224)))."""

    def __init__(self, featmap_names, output_size, sampling_ratio,
                 canonical_scale: int = 224, canonical_level: int = 4):
        super().__init__()
        self.featmap_names = featmap_names
        self.output_size = (output_size, output_size) if isinstance(output_size, int) else output_size
        self.sampling_ratio = sampling_ratio
        self.canonical_scale = canonical_scale
        self.canonical_level = canonical_level

    @staticmethod
    def _convert_to_roi_format(boxes_list):
        rois = []
        for i, b in enumerate(boxes_list):
            idx = torch.full((b.shape[0], 1), i, dtype=b.dtype, device=b.device)
            rois.append(torch.cat([idx, b], dim=1))
        return torch.cat(rois, dim=0)

    def forward(self, x: dict, boxes_list: list, image_shapes: list) -> torch.Tensor:
        feats = [x[k] for k in self.featmap_names if k in x]
        rois = self._convert_to_roi_format(boxes_list)
        # infer scales from feature/image size ratio
        orig_h = max(s[0] for s in image_shapes)
        scales = []
        for f in feats:
            scale = 2 ** int(round(torch.log2(torch.tensor(f.shape[-2] / orig_h)).item()))
            scales.append(float(scale))
        if len(feats) == 1:
            return roi_align(feats[0], rois, self.output_size, scales[0],
                             self.sampling_ratio, aligned=False)
        # map each roi to a level
        areas = (rois[:, 3] - rois[:, 1]) * (rois[:, 4] - rois[:, 2])
        k = torch.floor(self.canonical_level +
                        torch.log2(torch.sqrt(areas.clamp(min=1e-6)) / self.canonical_scale))
        lvl_min = -int(torch.log2(torch.tensor(scales[0])).item())
        lvl_max = -int(torch.log2(torch.tensor(scales[-1])).item())
        k = k.clamp(min=lvl_min, max=lvl_max).to(torch.int64) - lvl_min
        out = feats[0].new_zeros((rois.shape[0], feats[0].shape[1], *self.output_size))
        for lvl, (f, s) in enumerate(zip(feats, scales)):
            idx = torch.where(k == lvl)[0]
            if idx.numel() == 0:
                continue
            out[idx] = roi_align(f, rois[idx], self.output_size, s,
                                 self.sampling_ratio, aligned=False).to(out.dtype)
        return out
