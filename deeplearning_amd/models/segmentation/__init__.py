from . import deeplab, fcn, hrnet, sspnet, unet  # noqa: F401
from .hrnet import OhemCrossEntropy  # noqa: F401
