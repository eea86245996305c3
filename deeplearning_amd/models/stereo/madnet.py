"""MADNet: real-time self-adaptive stereo (pyramid encoder, correlation-based
disparity decoders, refinement, bilinear warping) + MAD online adaptation.

Reference parity: deep_stereo/MadNet (models/MadNet.py Pyramid_Encoder:18,
Disparity_Decoder:184, Refinement_Module:263, _linear_warping:509;
Stereo_Online_Adaptation.py:26-301) — re-designed: warping uses
F.grid_sample (HW-accelerated sampler); MAD module sampling trains one
randomly-chosen pyramid scale per step.
"""
from __future__ import annotations

import random

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..registry import register_model


def conv_lrelu(cin, cout, k=3, stride=1, dilation=1):
    return nn.Sequential(
        nn.Conv2d(cin, cout, k, stride, dilation * (k - 1) // 2,
                  dilation=dilation),
        nn.LeakyReLU(0.2, inplace=True))


class PyramidEncoder(nn.Module):
    """6-level feature pyramid, 1/2 .. 1/64 (ref MadNet.py:18)."""

    chans = [16, 32, 64, 96, 128, 192]

    def __init__(self):
        super().__init__()
        cin = 3
        self.levels = nn.ModuleList()
        for c in self.chans:
            self.levels.append(nn.Sequential(
                conv_lrelu(cin, c, stride=2), conv_lrelu(c, c)))
            cin = c

    def forward(self, x):
        feats = []
        for lvl in self.levels:
            x = lvl(x)
            feats.append(x)
        return feats  # [1/2, 1/4, 1/8, 1/16, 1/32, 1/64]


def linear_warp(feature: torch.Tensor, disp: torch.Tensor) -> torch.Tensor:
    """Warp right-image features left by disparity (ref _linear_warping:509)."""
    B, C, H, W = feature.shape
    xs = torch.linspace(-1, 1, W, device=feature.device)
    ys = torch.linspace(-1, 1, H, device=feature.device)
    grid_y, grid_x = torch.meshgrid(ys, xs, indexing="ij")
    grid_x = grid_x.unsqueeze(0) - 2 * disp.squeeze(1) / max(W - 1, 1)
    grid = torch.stack([grid_x, grid_y.unsqueeze(0).expand_as(grid_x)], dim=-1)
    return F.grid_sample(feature, grid, mode="bilinear", padding_mode="zeros",
                         align_corners=True)


def correlation(left: torch.Tensor, right: torch.Tensor, max_disp=2):
    """1-D cost volume over [-max_disp, max_disp] shifts."""
    B, C, H, W = left.shape
    costs = []
    for d in range(-max_disp, max_disp + 1):
        shifted = torch.roll(right, shifts=d, dims=3)
        costs.append((left * shifted).mean(1))
    return torch.stack(costs, dim=1)  # B, 2*max_disp+1, H, W


class DisparityDecoder(nn.Module):
    """Cost volume + features -> residual disparity (ref :184)."""

    def __init__(self, cin):
        super().__init__()
        self.net = nn.Sequential(
            conv_lrelu(cin, 128), conv_lrelu(128, 128), conv_lrelu(128, 96),
            conv_lrelu(96, 64), conv_lrelu(64, 32),
            nn.Conv2d(32, 1, 3, padding=1))

    def forward(self, x):
        return self.net(x)


class RefinementModule(nn.Module):
    """Dilated-conv residual refinement at 1/4 res (ref :263)."""

    def __init__(self, cin):
        super().__init__()
        self.net = nn.Sequential(
            conv_lrelu(cin, 128, dilation=1), conv_lrelu(128, 128, dilation=2),
            conv_lrelu(128, 128, dilation=4), conv_lrelu(128, 96, dilation=8),
            conv_lrelu(96, 64, dilation=16), conv_lrelu(64, 32, dilation=1),
            nn.Conv2d(32, 1, 3, padding=1))

    def forward(self, x):
        return self.net(x)


class MADNet(nn.Module):
    def __init__(self, max_disp=2):
        super().__init__()
        self.max_disp = max_disp
        self.encoder = PyramidEncoder()
        corr_c = 2 * max_disp + 1
        chans = PyramidEncoder.chans
        # decoders for levels 5..1 (1/64 .. 1/4)
        self.decoders = nn.ModuleList([
            DisparityDecoder(corr_c + chans[i] + (0 if i == 5 else 1))
            for i in range(5, 0, -1)])
        self.refine = RefinementModule(chans[1] + 1)

    def forward(self, left, right):
        lf = self.encoder(left)
        rf = self.encoder(right)
        disp = None
        disps = []
        for k, i in enumerate(range(5, 0, -1)):
            l_feat, r_feat = lf[i], rf[i]
            if disp is not None:
                up = F.interpolate(disp, size=l_feat.shape[-2:],
                                   mode="bilinear", align_corners=True) * 2.0
                r_feat = linear_warp(r_feat, up)
                x = torch.cat([correlation(l_feat, r_feat, self.max_disp),
                               l_feat, up], dim=1)
            else:
                x = torch.cat([correlation(l_feat, r_feat, self.max_disp),
                               l_feat], dim=1)
            res = self.decoders[k](x)
            disp = res if disp is None else res + up
            disps.append(disp)
        refined = disp + self.refine(torch.cat([lf[1], disp], dim=1))
        disps.append(refined)
        full = F.interpolate(refined, size=left.shape[-2:], mode="bilinear",
                             align_corners=True) * 4.0
        return full, disps


def reprojection_loss(left, right, disp):
    """Unsupervised photometric loss: warp right to left with disp."""
    warped = linear_warp(right, disp)
    return (warped - left).abs().mean()


class MADAdapter:
    """Modular ADaptation: each step trains ONE randomly-sampled pyramid
    portion via the photometric loss (ref Stereo_Online_Adaptation.py:26-301)."""

    def __init__(self, model: MADNet, lr=1e-4):
        self.model = model
        self.optims = [
            torch.optim.Adam(
                list(model.encoder.levels[5 - k].parameters()) +
                list(model.decoders[k].parameters()), lr=lr)
            for k in range(len(model.decoders))]

    def step(self, left, right):
        k = random.randrange(len(self.optims))
        full, disps = self.model(left, right)
        scale_disp = F.interpolate(disps[k], size=left.shape[-2:],
                                   mode="bilinear", align_corners=True)
        factor = left.shape[-1] / disps[k].shape[-1]
        loss = reprojection_loss(left, right, scale_disp * factor)
        opt = self.optims[k]
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        return float(loss.detach())


@register_model
def madnet(**kw):
    return MADNet(**kw)
