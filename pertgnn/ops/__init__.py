from . import functional, reference
from .backend import ext, has_hip

__all__ = ["functional", "reference", "ext", "has_hip"]
