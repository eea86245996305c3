from . import hrnet_pose  # noqa: F401
from .hrnet_pose import (KeypointToHeatMap, decode_heatmaps,  # noqa: F401
                         heatmap_focal_loss, heatmap_nms)
