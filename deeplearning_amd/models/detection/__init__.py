from . import (anchors, faster_rcnn, fcos, fpn, retinanet,  # noqa: F401
               roi_heads, rpn, transform, yolov5, yolox)
from .anchors import (AnchorGenerator, BalancedPositiveNegativeSampler,  # noqa: F401
                      BoxCoder, Matcher)
from .fpn import FeaturePyramidNetwork, resnet_fpn_backbone  # noqa: F401
from .transform import GeneralizedRCNNTransform, ImageList  # noqa: F401
from .yolov5 import ComputeLoss  # noqa: F401
from .yolox import simota_assign, yolox_postprocess  # noqa: F401
