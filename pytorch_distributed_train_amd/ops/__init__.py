from . import functional
from .backend import has_ext, ext

__all__ = ["functional", "has_ext", "ext"]
