from . import madnet  # noqa: F401
from .madnet import MADAdapter, linear_warp, reprojection_loss  # noqa: F401
