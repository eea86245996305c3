"""RepVGG with structural re-parameterization (train 3x3+1x1+identity branches,
deploy a single fused 3x3 conv).

Reference parity: classification/RepVGG/models/repvgg.py
(RepVGGBlock.get_equivalent_kernel_bias:93-133, switch_to_deploy:133) and
convert.py — re-designed on the framework's fused HIP BN+ReLU; the deploy-mode
fuse folds each branch's BN into conv weights exactly as the paper specifies.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import BatchNorm2d
from ..registry import register_model


def conv_bn(cin, cout, kernel_size, stride, padding, groups=1):
    return nn.Sequential(
        nn.Conv2d(cin, cout, kernel_size, stride, padding, groups=groups,
                  bias=False),
        BatchNorm2d(cout))


class RepVGGBlock(nn.Module):
    def __init__(self, cin, cout, stride=1, groups=1, deploy=False):
        super().__init__()
        self.deploy = deploy
        self.groups = groups
        self.in_channels = cin
        self.nonlinearity = nn.ReLU(inplace=True)
        if deploy:
            self.rbr_reparam = nn.Conv2d(cin, cout, 3, stride, 1, groups=groups,
                                         bias=True)
        else:
            self.rbr_identity = BatchNorm2d(cin) \
                if cout == cin and stride == 1 else None
            self.rbr_dense = conv_bn(cin, cout, 3, stride, 1, groups)
            self.rbr_1x1 = conv_bn(cin, cout, 1, stride, 0, groups)

    def forward(self, x):
        if self.deploy:
            return self.nonlinearity(self.rbr_reparam(x))
        out = self.rbr_dense(x) + self.rbr_1x1(x)
        if self.rbr_identity is not None:
            out = out + self.rbr_identity(x)
        return self.nonlinearity(out)

    # --- structural re-parameterization -----------------------------------
    def _fuse_bn(self, branch):
        if branch is None:
            return 0, 0
        if isinstance(branch, nn.Sequential):
            kernel = branch[0].weight
            bn = branch[1]
        else:  # identity BN branch: build an identity 3x3 kernel
            bn = branch
            input_dim = self.in_channels // self.groups
            kernel = torch.zeros(self.in_channels, input_dim, 3, 3,
                                 device=bn.weight.device, dtype=bn.weight.dtype)
            for i in range(self.in_channels):
                kernel[i, i % input_dim, 1, 1] = 1
        std = (bn.running_var + bn.eps).sqrt()
        t = (bn.weight / std).reshape(-1, 1, 1, 1)
        return kernel * t, bn.bias - bn.running_mean * bn.weight / std

    @staticmethod
    def _pad_1x1_to_3x3(k):
        if isinstance(k, int):
            return k
        return torch.nn.functional.pad(k, [1, 1, 1, 1])

    def get_equivalent_kernel_bias(self):
        k3, b3 = self._fuse_bn(self.rbr_dense)
        k1, b1 = self._fuse_bn(self.rbr_1x1)
        kid, bid = self._fuse_bn(self.rbr_identity)
        return k3 + self._pad_1x1_to_3x3(k1) + kid, b3 + b1 + bid

    @torch.no_grad()
    def switch_to_deploy(self):
        if self.deploy:
            return
        kernel, bias = self.get_equivalent_kernel_bias()
        dense = self.rbr_dense[0]
        self.rbr_reparam = nn.Conv2d(
            dense.in_channels, dense.out_channels, 3, dense.stride[0], 1,
            groups=dense.groups, bias=True)
        self.rbr_reparam.weight.copy_(kernel)
        self.rbr_reparam.bias.copy_(bias)
        del self.rbr_dense, self.rbr_1x1
        if hasattr(self, "rbr_identity"):
            del self.rbr_identity
        self.deploy = True


class RepVGG(nn.Module):
    def __init__(self, num_blocks, width_multiplier, num_classes=1000,
                 override_groups_map=None, deploy=False):
        super().__init__()
        self.deploy = deploy
        self.override_groups_map = override_groups_map or {}
        self.cur_layer_idx = 1

        self.in_planes = min(64, int(64 * width_multiplier[0]))
        self.stage0 = RepVGGBlock(3, self.in_planes, stride=2, deploy=deploy)
        self.stage1 = self._make_stage(int(64 * width_multiplier[0]), num_blocks[0])
        self.stage2 = self._make_stage(int(128 * width_multiplier[1]), num_blocks[1])
        self.stage3 = self._make_stage(int(256 * width_multiplier[2]), num_blocks[2])
        self.stage4 = self._make_stage(int(512 * width_multiplier[3]), num_blocks[3])
        self.gap = nn.AdaptiveAvgPool2d(1)
        self.linear = nn.Linear(int(512 * width_multiplier[3]), num_classes)

    def _make_stage(self, planes, num_blocks):
        strides = [2] + [1] * (num_blocks - 1)
        blocks = []
        for stride in strides:
            groups = self.override_groups_map.get(self.cur_layer_idx, 1)
            blocks.append(RepVGGBlock(self.in_planes, planes, stride,
                                      groups=groups, deploy=self.deploy))
            self.in_planes = planes
            self.cur_layer_idx += 1
        return nn.Sequential(*blocks)

    def forward(self, x):
        x = self.stage4(self.stage3(self.stage2(self.stage1(self.stage0(x)))))
        return self.linear(self.gap(x).flatten(1))


def repvgg_model_convert(model: nn.Module, save_path=None):
    """Fuse every RepVGGBlock to deploy form (ref convert.py semantics)."""
    for m in model.modules():
        if hasattr(m, "switch_to_deploy"):
            m.switch_to_deploy()
    if save_path is not None:
        torch.save(model.state_dict(), save_path)
    return model


_G2 = {l: 2 for l in [2, 4, 6, 8, 10, 12, 14, 16, 18, 20, 22, 24, 26]}


@register_model
def repvgg_a0(num_classes=1000, **kw):
    return RepVGG([2, 4, 14, 1], [0.75, 0.75, 0.75, 2.5],
                  num_classes=num_classes, **kw)


@register_model
def repvgg_a1(num_classes=1000, **kw):
    return RepVGG([2, 4, 14, 1], [1, 1, 1, 2.5], num_classes=num_classes, **kw)


@register_model
def repvgg_a2(num_classes=1000, **kw):
    return RepVGG([2, 4, 14, 1], [1.5, 1.5, 1.5, 2.75],
                  num_classes=num_classes, **kw)


@register_model
def repvgg_b0(num_classes=1000, **kw):
    return RepVGG([4, 6, 16, 1], [1, 1, 1, 2.5], num_classes=num_classes, **kw)


@register_model
def repvgg_b1(num_classes=1000, **kw):
    return RepVGG([4, 6, 16, 1], [2, 2, 2, 4], num_classes=num_classes, **kw)


@register_model
def repvgg_b1g2(num_classes=1000, **kw):
    return RepVGG([4, 6, 16, 1], [2, 2, 2, 4], num_classes=num_classes,
                  override_groups_map=_G2, **kw)
