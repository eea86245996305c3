from . import mae, supcon  # noqa: F401
from .supcon import SupConLoss  # noqa: F401
