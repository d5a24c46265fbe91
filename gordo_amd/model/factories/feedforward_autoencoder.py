"""Feedforward autoencoder factories.

Parity: gordo_components/model/factories/feedforward_autoencoder.py —
``feedforward_model`` (explicit encode/decode dims), ``feedforward_symmetric``
(mirrored dims), ``feedforward_hourglass`` (compression_factor +
encoding_layers → hourglass_calc_dims). Defaults mirror the reference:
tanh hidden activations, linear output, Adam + MSE (consumed by the trainer).
"""
from __future__ import annotations

from typing import Sequence

from ..register import register_model_builder
from ..torch_nets import FeedForwardNet
from .utils import check_dim_func_len, hourglass_calc_dims


@register_model_builder(type="TorchAutoEncoder")
def feedforward_model(
    n_features: int,
    n_features_out: int = None,
    encoding_dim: Sequence[int] = (256, 128, 64),
    encoding_func: Sequence[str] = None,
    decoding_dim: Sequence[int] = (64, 128, 256),
    decoding_func: Sequence[str] = None,
    out_func: str = "linear",
    **kwargs,
) -> FeedForwardNet:
    n_features_out = n_features_out or n_features
    encoding_func = list(encoding_func) if encoding_func else ["tanh"] * len(encoding_dim)
    decoding_func = list(decoding_func) if decoding_func else ["tanh"] * len(decoding_dim)
    check_dim_func_len("encoding", encoding_dim, encoding_func)
    check_dim_func_len("decoding", decoding_dim, decoding_func)
    dims = list(encoding_dim) + list(decoding_dim)
    funcs = encoding_func + decoding_func
    return FeedForwardNet(n_features, n_features_out, dims, funcs, out_func)


@register_model_builder(type="TorchAutoEncoder")
def feedforward_symmetric(
    n_features: int,
    n_features_out: int = None,
    dims: Sequence[int] = (256, 128, 64),
    funcs: Sequence[str] = None,
    out_func: str = "linear",
    **kwargs,
) -> FeedForwardNet:
    if not dims:
        raise ValueError("dims must be a non-empty list of layer sizes")
    funcs = list(funcs) if funcs else ["tanh"] * len(dims)
    check_dim_func_len("", dims, funcs)
    return feedforward_model(
        n_features,
        n_features_out,
        encoding_dim=list(dims),
        encoding_func=list(funcs),
        decoding_dim=list(dims[::-1]),
        decoding_func=list(funcs[::-1]),
        out_func=out_func,
        **kwargs,
    )


@register_model_builder(type="TorchAutoEncoder")
def feedforward_hourglass(
    n_features: int,
    n_features_out: int = None,
    encoding_layers: int = 3,
    compression_factor: float = 0.5,
    func: str = "tanh",
    out_func: str = "linear",
    **kwargs,
) -> FeedForwardNet:
    """Hourglass: dims interpolate n_features → n_features*compression_factor
    over ``encoding_layers`` layers, then mirror back up.

    Matches the reference composition hourglass == symmetric(hourglass_calc_dims)
    — the bottleneck layer is duplicated, e.g. n_features=50, encoding_layers=2,
    compression=0.5 → hidden stack 38→25→25→38 (i.e. 50→38→25→25→38→50 with the
    input/output widths).
    """
    dims = hourglass_calc_dims(compression_factor, encoding_layers, n_features)
    return feedforward_symmetric(
        n_features,
        n_features_out,
        dims=dims,
        funcs=[func] * len(dims),
        out_func=out_func,
        **kwargs,
    )
