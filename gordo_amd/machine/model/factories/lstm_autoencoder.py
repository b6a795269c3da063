"""
LSTM autoencoder / forecast model factories.

Mirror of the reference Keras builders
(gordo/machine/model/factories/lstm_autoencoder.py:17-263), emitting an
engine ``ModelSpec`` consumed by the grouped LSTM device engine
(engine/pack.LSTMPack). Semantics kept: stacked LSTM encoder with
return_sequences=True, decoder whose LAST layer has
return_sequences=False, Dense output layer, loss=mse, Adam.
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

from ..register import register_model_builder
from .utils import check_dim_func_len, hourglass_calc_dims
from ....engine.spec import LayerSpec, ModelSpec


@register_model_builder(type="KerasLSTMAutoEncoder")
@register_model_builder(type="KerasLSTMForecast")
def lstm_model(
    n_features: int,
    n_features_out: Optional[int] = None,
    lookback_window: int = 1,
    encoding_dim: Tuple[int, ...] = (256, 128, 64),
    encoding_func: Tuple[str, ...] = ("tanh", "tanh", "tanh"),
    decoding_dim: Tuple[int, ...] = (64, 128, 256),
    decoding_func: Tuple[str, ...] = ("tanh", "tanh", "tanh"),
    out_func: str = "linear",
    optimizer: str = "Adam",
    optimizer_kwargs: Optional[Dict[str, Any]] = None,
    compile_kwargs: Optional[Dict[str, Any]] = None,
    **kwargs,
) -> ModelSpec:
    """
    >>> spec = lstm_model(10, lookback_window=4, encoding_dim=(8,),
    ...                   encoding_func=("tanh",), decoding_dim=(8,),
    ...                   decoding_func=("tanh",))
    >>> [(l.kind, l.units, l.return_sequences) for l in spec.layers]
    [('lstm', 8, True), ('lstm', 8, False), ('dense', 10, True)]
    """
    n_features_out = n_features_out or n_features
    check_dim_func_len("encoding", encoding_dim, encoding_func)
    check_dim_func_len("decoding", decoding_dim, decoding_func)

    layers = []
    for units, activation in zip(encoding_dim, encoding_func):
        layers.append(
            LayerSpec(
                kind="lstm",
                units=int(units),
                activation=activation,
                return_sequences=True,
            )
        )
    for i, (units, activation) in enumerate(zip(decoding_dim, decoding_func)):
        layers.append(
            LayerSpec(
                kind="lstm",
                units=int(units),
                activation=activation,
                return_sequences=i != len(decoding_dim) - 1,
            )
        )
    layers.append(
        LayerSpec(kind="dense", units=int(n_features_out), activation=out_func)
    )
    return ModelSpec(
        model_type="lstm",
        n_features=int(n_features),
        n_features_out=int(n_features_out),
        layers=layers,
        lookback_window=int(lookback_window),
        loss="mse",
        optimizer=optimizer if isinstance(optimizer, str) else "Adam",
        optimizer_kwargs=dict(optimizer_kwargs or {}),
    )


@register_model_builder(type="KerasLSTMAutoEncoder")
@register_model_builder(type="KerasLSTMForecast")
def lstm_symmetric(
    n_features: int,
    n_features_out: Optional[int] = None,
    lookback_window: int = 1,
    dims: Tuple[int, ...] = (256, 128, 64),
    funcs: Tuple[str, ...] = ("tanh", "tanh", "tanh"),
    out_func: str = "linear",
    optimizer: str = "Adam",
    optimizer_kwargs: Optional[Dict[str, Any]] = None,
    compile_kwargs: Optional[Dict[str, Any]] = None,
    **kwargs,
) -> ModelSpec:
    """
    >>> spec = lstm_symmetric(10, dims=(8, 4), funcs=("tanh", "tanh"))
    >>> [(l.units, l.return_sequences) for l in spec.layers if l.kind == "lstm"]
    [(8, True), (4, True), (4, True), (8, False)]
    """
    if len(dims) == 0:
        raise ValueError("Parameter dims must have len > 0")
    if len(dims) != len(funcs):
        raise ValueError("Length of dims and funcs must be equal")
    return lstm_model(
        n_features,
        n_features_out,
        lookback_window=lookback_window,
        encoding_dim=tuple(dims),
        encoding_func=tuple(funcs),
        decoding_dim=tuple(reversed(dims)),
        decoding_func=tuple(reversed(funcs)),
        out_func=out_func,
        optimizer=optimizer,
        optimizer_kwargs=optimizer_kwargs,
        compile_kwargs=compile_kwargs,
        **kwargs,
    )


@register_model_builder(type="KerasLSTMAutoEncoder")
@register_model_builder(type="KerasLSTMForecast")
def lstm_hourglass(
    n_features: int,
    n_features_out: Optional[int] = None,
    lookback_window: int = 1,
    encoding_layers: int = 3,
    compression_factor: float = 0.5,
    func: str = "tanh",
    out_func: str = "linear",
    optimizer: str = "Adam",
    optimizer_kwargs: Optional[Dict[str, Any]] = None,
    compile_kwargs: Optional[Dict[str, Any]] = None,
    **kwargs,
) -> ModelSpec:
    """
    >>> spec = lstm_hourglass(10)
    >>> [l.units for l in spec.layers if l.kind == "lstm"]
    [8, 7, 5, 5, 7, 8]
    """
    dims = hourglass_calc_dims(compression_factor, encoding_layers, n_features)
    return lstm_symmetric(
        n_features,
        n_features_out,
        lookback_window=lookback_window,
        dims=dims,
        funcs=tuple([func] * len(dims)),
        out_func=out_func,
        optimizer=optimizer,
        optimizer_kwargs=optimizer_kwargs,
        compile_kwargs=compile_kwargs,
        **kwargs,
    )
