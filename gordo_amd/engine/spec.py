"""
ModelSpec — the device-engine model description.

The factory functions in ``machine/model/factories`` (the analog of the
reference's Keras model builders, feedforward_autoencoder.py /
lstm_autoencoder.py) produce these specs; the packed engine
(``engine/pack.py``) materializes G of them at a time on one GPU.
"""
from __future__ import annotations

from dataclasses import dataclass, field, asdict
from typing import Any, Dict, List


@dataclass
class LayerSpec:
    kind: str  # "dense" | "lstm"
    units: int
    activation: str = "tanh"
    l1_activity: float = 0.0  # L1 activity regularizer weight (dense)
    return_sequences: bool = True  # lstm only

    def to_dict(self) -> Dict[str, Any]:
        return asdict(self)

    @classmethod
    def from_dict(cls, d) -> "LayerSpec":
        return cls(**d)


@dataclass
class ModelSpec:
    model_type: str  # "feedforward" | "lstm"
    n_features: int
    n_features_out: int
    layers: List[LayerSpec] = field(default_factory=list)
    lookback_window: int = 1  # lstm path
    lookahead: int = 0  # 0 = autoencoder, 1 = one-step forecast
    loss: str = "mse"
    optimizer: str = "Adam"
    optimizer_kwargs: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d = asdict(self)
        return d

    @classmethod
    def from_dict(cls, d) -> "ModelSpec":
        d = dict(d)
        d["layers"] = [LayerSpec.from_dict(x) for x in d.get("layers", [])]
        return cls(**d)

    # ---- shape helpers -------------------------------------------------
    def dense_dims(self) -> List[tuple]:
        """[(in, out, act, l1), ...] for a pure-dense spec."""
        assert self.model_type == "feedforward"
        dims = []
        prev = self.n_features
        for layer in self.layers:
            dims.append((prev, layer.units, layer.activation, layer.l1_activity))
            prev = layer.units
        return dims

    def arch_key(self) -> tuple:
        """Hashable architecture key — models with equal keys can share
        one pack (grouped-GEMM batch)."""
        return (
            self.model_type,
            self.n_features,
            self.n_features_out,
            tuple(
                (l.kind, l.units, l.activation, l.l1_activity, l.return_sequences)
                for l in self.layers
            ),
            self.lookback_window,
            self.lookahead,
            self.loss,
            self.optimizer,
            tuple(sorted(self.optimizer_kwargs.items())),
        )

    @property
    def adam_params(self):
        kw = self.optimizer_kwargs or {}
        return dict(
            lr=float(kw.get("lr", kw.get("learning_rate", 0.001))),
            beta1=float(kw.get("beta_1", 0.9)),
            beta2=float(kw.get("beta_2", 0.999)),
            eps=float(kw.get("epsilon", 1e-7)),  # keras default epsilon
        )
