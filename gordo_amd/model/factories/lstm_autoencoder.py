"""LSTM autoencoder factories.

Parity: gordo_components/model/factories/lstm_autoencoder.py — ``lstm_model``,
``lstm_symmetric``, ``lstm_hourglass``; stacked LSTM layers over
``(batch, lookback_window, n_features)`` with a dense output head; hourglass
dim logic shared with the feedforward factories via factories/utils.py.
"""
from __future__ import annotations

from typing import Sequence

from ..register import register_model_builder
from ..torch_nets import LSTMNet
from .utils import check_dim_func_len, hourglass_calc_dims


@register_model_builder(type="TorchLSTMAutoEncoder")
@register_model_builder(type="TorchLSTMForecast")
def lstm_model(
    n_features: int,
    n_features_out: int = None,
    lookback_window: int = 144,
    encoding_dim: Sequence[int] = (256, 128, 64),
    encoding_func: Sequence[str] = None,
    decoding_dim: Sequence[int] = (64, 128, 256),
    decoding_func: Sequence[str] = None,
    out_func: str = "linear",
    **kwargs,
) -> LSTMNet:
    n_features_out = n_features_out or n_features
    encoding_func = list(encoding_func) if encoding_func else ["tanh"] * len(encoding_dim)
    decoding_func = list(decoding_func) if decoding_func else ["tanh"] * len(decoding_dim)
    check_dim_func_len("encoding", encoding_dim, encoding_func)
    check_dim_func_len("decoding", decoding_dim, decoding_func)
    dims = list(encoding_dim) + list(decoding_dim)
    funcs = encoding_func + decoding_func
    return LSTMNet(
        n_features,
        n_features_out,
        dims,
        funcs,
        out_func=out_func,
        lookback_window=lookback_window,
    )


@register_model_builder(type="TorchLSTMAutoEncoder")
@register_model_builder(type="TorchLSTMForecast")
def lstm_symmetric(
    n_features: int,
    n_features_out: int = None,
    lookback_window: int = 144,
    dims: Sequence[int] = (256, 128, 64),
    funcs: Sequence[str] = None,
    out_func: str = "linear",
    **kwargs,
) -> LSTMNet:
    if not dims:
        raise ValueError("dims must be a non-empty list of layer sizes")
    funcs = list(funcs) if funcs else ["tanh"] * len(dims)
    check_dim_func_len("", dims, funcs)
    return lstm_model(
        n_features,
        n_features_out,
        lookback_window=lookback_window,
        encoding_dim=list(dims),
        encoding_func=list(funcs),
        decoding_dim=list(dims[::-1]),
        decoding_func=list(funcs[::-1]),
        out_func=out_func,
        **kwargs,
    )


@register_model_builder(type="TorchLSTMAutoEncoder")
@register_model_builder(type="TorchLSTMForecast")
def lstm_hourglass(
    n_features: int,
    n_features_out: int = None,
    lookback_window: int = 144,
    encoding_layers: int = 3,
    compression_factor: float = 0.5,
    func: str = "tanh",
    out_func: str = "linear",
    **kwargs,
) -> LSTMNet:
    # Reference composition: hourglass == symmetric(hourglass_calc_dims) —
    # the bottleneck layer is duplicated (e.g. 50 tags, 2 layers, 0.5 →
    # LSTM stack 38→25→25→38), consistent with feedforward_hourglass.
    dims = hourglass_calc_dims(compression_factor, encoding_layers, n_features)
    return lstm_symmetric(
        n_features,
        n_features_out,
        lookback_window=lookback_window,
        dims=dims,
        funcs=[func] * len(dims),
        out_func=out_func,
        **kwargs,
    )
