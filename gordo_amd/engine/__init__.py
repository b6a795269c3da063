from .spec import ModelSpec, LayerSpec
from .pack import DensePack, LSTMPack, BasePack

__all__ = ["ModelSpec", "LayerSpec", "DensePack", "LSTMPack", "BasePack"]
