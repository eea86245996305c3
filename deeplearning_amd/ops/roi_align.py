"""RoIAlign (torchvision semantics) + MultiScaleRoIAlign (FPN level mapper).

Reference parity: torchvision.ops.MultiScaleRoIAlign in fasterRcnn
(models/faster_rcnn.py:8,305-309). GPU path: csrc/roialign.hip; CPU reference
is a pure-PyTorch bilinear implementation used by the parity tests.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import ext, use_hip


def _roi_align_group(input, rois, output_size, spatial_scale, gh, gw, aligned,
                     chunk=64):
    """Vectorized RoIAlign for a group of rois sharing one (gh, gw) sampling
    grid. Exact torchvision semantics (count = gh*gw, out-of-range samples
    contribute 0, coords clamped into [0, dim-1])."""
    PH, PW = output_size
    N, C, H, W = input.shape
    R = rois.shape[0]
    off = 0.5 if aligned else 0.0
    b_idx = rois[:, 0].long()
    x1 = rois[:, 1] * spatial_scale - off
    y1 = rois[:, 2] * spatial_scale - off
    x2 = rois[:, 3] * spatial_scale - off
    y2 = rois[:, 4] * spatial_scale - off
    rw, rh = x2 - x1, y2 - y1
    if not aligned:
        rw, rh = rw.clamp(min=1.0), rh.clamp(min=1.0)
    bh, bw = rh / PH, rw / PW

    ph = torch.arange(PH, dtype=input.dtype, device=input.device)
    pw = torch.arange(PW, dtype=input.dtype, device=input.device)
    iy = torch.arange(gh, dtype=input.dtype, device=input.device)
    ix = torch.arange(gw, dtype=input.dtype, device=input.device)
    # ys: [R, PH, gh], xs: [R, PW, gw]
    ys = y1[:, None, None] + ph[None, :, None] * bh[:, None, None] + \
        (iy[None, None, :] + 0.5) * (bh[:, None, None] / gh)
    xs = x1[:, None, None] + pw[None, :, None] * bw[:, None, None] + \
        (ix[None, None, :] + 0.5) * (bw[:, None, None] / gw)

    out = input.new_empty((R, C, PH, PW))
    flat = input.reshape(N, C, H * W)
    S = PH * gh * PW * gw
    for s in range(0, R, chunk):
        e = min(s + chunk, R)
        y = ys[s:e].reshape(e - s, PH * gh, 1, 1)
        x = xs[s:e].reshape(e - s, 1, 1, PW * gw)
        y = y.expand(e - s, PH * gh, 1, PW * gw)
        x = x.expand(e - s, PH * gh, 1, PW * gw)
        valid = (y >= -1.0) & (y <= H) & (x >= -1.0) & (x <= W)
        yy = y.clamp(min=0.0)
        xx = x.clamp(min=0.0)
        y0 = yy.floor().long().clamp(max=H - 1)
        x0 = xx.floor().long().clamp(max=W - 1)
        hi_y = (y0 >= H - 1)
        hi_x = (x0 >= W - 1)
        y1i = (y0 + 1).clamp(max=H - 1)
        x1i = (x0 + 1).clamp(max=W - 1)
        yy = torch.where(hi_y, y0.to(yy.dtype), yy)
        xx = torch.where(hi_x, x0.to(xx.dtype), xx)
        ly, lx = yy - y0, xx - x0
        hy, hx = 1 - ly, 1 - lx
        w00 = (hy * hx * valid).reshape(e - s, 1, -1)
        w01 = (hy * lx * valid).reshape(e - s, 1, -1)
        w10 = (ly * hx * valid).reshape(e - s, 1, -1)
        w11 = (ly * lx * valid).reshape(e - s, 1, -1)
        i00 = (y0 * W + x0).reshape(e - s, 1, -1).expand(-1, C, -1)
        i01 = (y0 * W + x1i).reshape(e - s, 1, -1).expand(-1, C, -1)
        i10 = (y1i * W + x0).reshape(e - s, 1, -1).expand(-1, C, -1)
        i11 = (y1i * W + x1i).reshape(e - s, 1, -1).expand(-1, C, -1)
        src = flat[b_idx[s:e]]  # [r, C, H*W]
        acc = w00 * src.gather(2, i00) + w01 * src.gather(2, i01) + \
            w10 * src.gather(2, i10) + w11 * src.gather(2, i11)
        acc = acc.reshape(e - s, C, PH, gh, PW, gw).sum(dim=(3, 5))
        out[s:e] = acc / (gh * gw)
    return out


def _roi_align_eager(input, rois, output_size, spatial_scale, sampling_ratio,
                     aligned):
    """Pure-PyTorch RoIAlign (exact semantics; CPU reference). Vectorized per
    (gh, gw) sampling-grid group."""
    PH, PW = output_size
    R = rois.shape[0]
    if R == 0:
        return input.new_zeros((0, input.shape[1], PH, PW))
    off = 0.5 if aligned else 0.0
    rh = (rois[:, 4] - rois[:, 2]) * spatial_scale
    rw = (rois[:, 3] - rois[:, 1]) * spatial_scale
    if not aligned:
        rh, rw = rh.clamp(min=1.0), rw.clamp(min=1.0)
    if sampling_ratio > 0:
        gh = torch.full((R,), sampling_ratio, dtype=torch.long)
        gw = gh
    else:
        gh = (rh / PH).ceil().long().clamp(min=1)
        gw = (rw / PW).ceil().long().clamp(min=1)
    out = input.new_zeros((R, input.shape[1], PH, PW))
    key = gh * 10000 + gw
    for k in key.unique():
        idx = torch.where(key == k)[0]
        out[idx] = _roi_align_group(
            input, rois[idx], output_size, spatial_scale,
            int(gh[idx[0]]), int(gw[idx[0]]), aligned)
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
