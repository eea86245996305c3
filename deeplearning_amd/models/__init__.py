from .registry import build_model, list_models, register_model

# importing submodules registers their factories
from .classification import lenet, resnet, vit  # noqa: F401

__all__ = ["build_model", "list_models", "register_model"]
