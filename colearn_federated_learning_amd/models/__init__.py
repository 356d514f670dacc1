from .base import ColearnModel  # noqa: F401
from .registry import build_model  # noqa: F401
