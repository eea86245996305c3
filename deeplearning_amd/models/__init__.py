from .registry import build_model, list_models, register_model

# importing submodules registers their factories
from . import classification  # noqa: F401,E402
from . import detection  # noqa: F401,E402
from . import metric  # noqa: F401,E402
from . import pose  # noqa: F401,E402
from . import segmentation  # noqa: F401,E402
from . import ssl  # noqa: F401,E402
from . import stereo  # noqa: F401,E402

__all__ = ["build_model", "list_models", "register_model"]
