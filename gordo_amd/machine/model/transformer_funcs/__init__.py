from . import general

__all__ = ["general"]
