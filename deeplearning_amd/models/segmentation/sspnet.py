"""SSP few-shot segmentation: self-support prototypes over a frozen ResNet-50.

Reference parity: Image_segmentation/few_shot_segmentation/models/sspnet.py
(SSPNet:8-136, masked avg-pool :119, self-support prototypes :136) —
re-designed on this repo's ResNet features + cosine-similarity matching.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..classification.resnet import ResNet, Bottleneck
from ..registry import register_model


def masked_average_pooling(feature, mask):
    """feature: B,C,h,w ; mask: B,H,W in {0,1} -> B,C prototype."""
    mask = F.interpolate(mask.unsqueeze(1), size=feature.shape[-2:],
                         mode="bilinear", align_corners=True)
    return (feature * mask).sum(dim=(2, 3)) / (mask.sum(dim=(2, 3)) + 1e-5)


class SSPNet(nn.Module):
    def __init__(self, refine=True):
        super().__init__()
        backbone = ResNet(Bottleneck, [3, 4, 6, 3], include_top=False)
        self.layer0 = nn.Sequential(backbone.conv1, backbone.bn1,
                                    backbone.maxpool)
        self.layer1, self.layer2, self.layer3 = (backbone.layer1,
                                                 backbone.layer2,
                                                 backbone.layer3)
        self.refine = refine

    def extract(self, x):
        return self.layer3(self.layer2(self.layer1(self.layer0(x))))

    @staticmethod
    def similarity(feature, fg_proto, bg_proto):
        f = F.normalize(feature, dim=1)
        fg = F.normalize(fg_proto, dim=1)[..., None, None]
        bg = F.normalize(bg_proto, dim=1)[..., None, None]
        sim_fg = (f * fg).sum(1)
        sim_bg = (f * bg).sum(1)
        return torch.stack([sim_bg, sim_fg], dim=1) * 10.0

    def self_support_prototype(self, query_feat, pred):
        """Use confident query pixels as additional (self-support) prototypes."""
        prob = pred.softmax(1)
        fg_mask = (prob[:, 1] > 0.7).float()
        bg_mask = (prob[:, 0] > 0.7).float()
        C = query_feat.shape[1]
        fg = (query_feat * fg_mask.unsqueeze(1)).sum((2, 3)) / \
            (fg_mask.sum((1, 2)).unsqueeze(1) + 1e-5)
        bg = (query_feat * bg_mask.unsqueeze(1)).sum((2, 3)) / \
            (bg_mask.sum((1, 2)).unsqueeze(1) + 1e-5)
        return fg, bg

    def forward(self, support_img, support_mask, query_img):
        with torch.no_grad() if not self.training else torch.enable_grad():
            s_feat = self.extract(support_img)
            q_feat = self.extract(query_img)
        fg_proto = masked_average_pooling(s_feat, support_mask.float())
        bg_proto = masked_average_pooling(s_feat, 1 - support_mask.float())
        pred = self.similarity(q_feat, fg_proto, bg_proto)
        if self.refine:
            ss_fg, ss_bg = self.self_support_prototype(q_feat, pred)
            fg2 = 0.5 * fg_proto + 0.5 * ss_fg
            bg2 = 0.3 * bg_proto + 0.7 * ss_bg
            pred = self.similarity(q_feat, fg2, bg2)
        return F.interpolate(pred, size=query_img.shape[-2:], mode="bilinear",
                             align_corners=True)


@register_model
def sspnet(**kw):
    return SSPNet(**kw)
