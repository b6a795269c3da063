from . import disk_registry
from .utils import capture_args
from .text import replace_all_non_ascii_chars

__all__ = ["disk_registry", "capture_args", "replace_all_non_ascii_chars"]
