"""
Feedforward autoencoder model factories.

These mirror the reference Keras builders
(gordo/machine/model/factories/feedforward_autoencoder.py) but emit an
engine ``ModelSpec`` consumed by the grouped MFMA device engine
(engine/pack.DensePack) instead of a Keras graph. Semantics kept:
encoder/decoder dims+activations, l1(1e-4) activity regularizer on
every encoder layer except the first (reference :81), linear output
layer, MSE loss, Adam.
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

from ..register import register_model_builder
from .utils import check_dim_func_len, hourglass_calc_dims
from ....engine.spec import LayerSpec, ModelSpec

L1_ACTIVITY = 10e-5  # the reference's regularizers.l1(10e-5)


@register_model_builder(type="KerasAutoEncoder")
def feedforward_model(
    n_features: int,
    n_features_out: Optional[int] = None,
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
    Build a feedforward autoencoder spec.

    >>> spec = feedforward_model(10, encoding_dim=(8, 5), encoding_func=("tanh","tanh"),
    ...                          decoding_dim=(5, 8), decoding_func=("tanh","tanh"))
    >>> [l.units for l in spec.layers]
    [8, 5, 5, 8, 10]
    >>> [l.l1_activity for l in spec.layers]
    [0.0, 0.0001, 0.0, 0.0, 0.0]
    """
    n_features_out = n_features_out or n_features
    check_dim_func_len("encoding", encoding_dim, encoding_func)
    check_dim_func_len("decoding", decoding_dim, decoding_func)

    layers = []
    for i, (units, activation) in enumerate(zip(encoding_dim, encoding_func)):
        layers.append(
            LayerSpec(
                kind="dense",
                units=int(units),
                activation=activation,
                l1_activity=0.0 if i == 0 else L1_ACTIVITY,
            )
        )
    for units, activation in zip(decoding_dim, decoding_func):
        layers.append(LayerSpec(kind="dense", units=int(units), activation=activation))
    layers.append(LayerSpec(kind="dense", units=int(n_features_out), activation=out_func))

    return ModelSpec(
        model_type="feedforward",
        n_features=int(n_features),
        n_features_out=int(n_features_out),
        layers=layers,
        loss="mse",
        optimizer=optimizer if isinstance(optimizer, str) else "Adam",
        optimizer_kwargs=dict(optimizer_kwargs or {}),
    )


@register_model_builder(type="KerasAutoEncoder")
def feedforward_symmetric(
    n_features: int,
    n_features_out: Optional[int] = None,
    dims: Tuple[int, ...] = (256, 128, 64),
    funcs: Tuple[str, ...] = ("tanh", "tanh", "tanh"),
    optimizer: str = "Adam",
    optimizer_kwargs: Optional[Dict[str, Any]] = None,
    compile_kwargs: Optional[Dict[str, Any]] = None,
    **kwargs,
) -> ModelSpec:
    """
    Symmetric autoencoder: encoder = dims, decoder = reversed(dims).

    >>> spec = feedforward_symmetric(10, dims=(8, 5), funcs=("tanh", "tanh"))
    >>> [l.units for l in spec.layers]
    [8, 5, 5, 8, 10]
    """
    if len(dims) == 0:
        raise ValueError("Parameter dims must have len > 0")
    if len(dims) != len(funcs):
        raise ValueError("Length of dims and funcs must be equal")
    return feedforward_model(
        n_features,
        n_features_out,
        encoding_dim=tuple(dims),
        encoding_func=tuple(funcs),
        decoding_dim=tuple(reversed(dims)),
        decoding_func=tuple(reversed(funcs)),
        optimizer=optimizer,
        optimizer_kwargs=optimizer_kwargs,
        compile_kwargs=compile_kwargs,
        **kwargs,
    )


@register_model_builder(type="KerasAutoEncoder")
def feedforward_hourglass(
    n_features: int,
    n_features_out: Optional[int] = None,
    encoding_layers: int = 3,
    compression_factor: float = 0.5,
    func: str = "tanh",
    optimizer: str = "Adam",
    optimizer_kwargs: Optional[Dict[str, Any]] = None,
    compile_kwargs: Optional[Dict[str, Any]] = None,
    **kwargs,
) -> ModelSpec:
    """
    Hourglass autoencoder: layer sizes interpolate from n_features down
    to compression_factor * n_features and back.

    >>> spec = feedforward_hourglass(10)
    >>> [l.units for l in spec.layers]
    [8, 7, 5, 5, 7, 8, 10]
    >>> spec = feedforward_hourglass(5, compression_factor=0.2)
    >>> [l.units for l in spec.layers]
    [4, 2, 1, 1, 2, 4, 5]
    """
    dims = hourglass_calc_dims(compression_factor, encoding_layers, n_features)
    return feedforward_symmetric(
        n_features,
        n_features_out,
        dims=dims,
        funcs=tuple([func] * len(dims)),
        optimizer=optimizer,
        optimizer_kwargs=optimizer_kwargs,
        compile_kwargs=compile_kwargs,
        **kwargs,
    )
