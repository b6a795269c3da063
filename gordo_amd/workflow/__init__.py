from .config_elements.normalized_config import NormalizedConfig

__all__ = ["NormalizedConfig"]
