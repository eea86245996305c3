"""Faster R-CNN: transform -> ResNet50-FPN -> RPN -> RoIHeads.

Reference parity: detection/fasterRcnn/models/faster_rcnn.py
(FasterRCNNBase.forward:44-100, FasterRCNN:305) — re-designed from this
repo's shared detection components.
"""
from __future__ import annotations

from ..registry import register_model
from torch import nn

from .anchors import AnchorGenerator
from .fpn import resnet_fpn_backbone
from .roi_heads import RoIHeads
from .rpn import RegionProposalNetwork, RPNHead
from .transform import GeneralizedRCNNTransform


class FasterRCNN(nn.Module):
    def __init__(self, num_classes=91, min_size=800, max_size=1333,
                 trainable_backbone_layers=3, **roi_kw):
        super().__init__()
        self.transform = GeneralizedRCNNTransform(min_size, max_size)
        self.backbone = resnet_fpn_backbone(
            trainable_layers=trainable_backbone_layers,
            extra_blocks="maxpool")
        anchor_gen = AnchorGenerator(
            sizes=((32,), (64,), (128,), (256,), (512,)),
            aspect_ratios=((0.5, 1.0, 2.0),) * 5)
        head = RPNHead(256, anchor_gen.num_anchors_per_location()[0])
        self.rpn = RegionProposalNetwork(anchor_gen, head)
        self.roi_heads = RoIHeads(num_classes, **roi_kw)

    def forward(self, images, targets=None):
        original_sizes = [tuple(img.shape[-2:]) for img in images]
        image_list, targets = self.transform(images, targets)
        features = self.backbone(image_list.tensors)
        proposals, rpn_losses = self.rpn(image_list, features, targets)
        detections, roi_losses = self.roi_heads(
            features, proposals, image_list.image_sizes, targets)
        if self.training:
            return {**rpn_losses, **roi_losses}
        return self.transform.postprocess(
            detections, image_list.image_sizes, original_sizes)


@register_model
def fasterrcnn_resnet50_fpn(num_classes=91, **kw):
    return FasterRCNN(num_classes=num_classes, **kw)
