from ._ext import ext, has_ext, use_hip
from .activations import GELU, SiLU, add_relu, gelu, silu
from .batchnorm import BatchNorm2d, FrozenBatchNorm2d
from .boxes import (batched_nms, bbox_iou_aligned, box_area, box_iou,
                    clip_boxes_to_image, generalized_box_iou, nms,
                    remove_small_boxes)
from .drop import DropPath, drop_path
from .ema import ModelEMA
from .layernorm import LayerNorm, LayerNorm2d, layer_norm
from .losses import (CrossEntropyLoss, cross_entropy, sigmoid_focal_loss,
                     soft_target_cross_entropy)
from .roi_align import MultiScaleRoIAlign, roi_align
from .window import (roll_and_window_partition, window_merge_and_roll,
                     window_partition_eager, window_reverse_eager)

__all__ = [
    "ext", "has_ext", "use_hip",
    "gelu", "silu", "add_relu", "GELU", "SiLU",
    "BatchNorm2d", "FrozenBatchNorm2d",
    "box_iou", "generalized_box_iou", "bbox_iou_aligned", "box_area", "nms",
    "batched_nms", "clip_boxes_to_image", "remove_small_boxes",
    "DropPath", "drop_path", "ModelEMA",
    "LayerNorm", "LayerNorm2d", "layer_norm",
    "CrossEntropyLoss", "cross_entropy", "soft_target_cross_entropy",
    "sigmoid_focal_loss",
    "MultiScaleRoIAlign", "roi_align",
    "roll_and_window_partition", "window_merge_and_roll",
    "window_partition_eager", "window_reverse_eager",
]
