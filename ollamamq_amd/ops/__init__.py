from . import interface
from .interface import AttnMeta

__all__ = ["interface", "AttnMeta"]
