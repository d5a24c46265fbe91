"""Torch network definitions produced by the factories.

These are thin, introspectable graphs: the fused MI355X engine
(gordo_amd.ops) reads ``dims``/``funcs`` off FeedForwardNet and runs the whole
forward+backward in hand-written HIP kernels; the modules themselves are the
correctness reference and the CPU path.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch
from torch import nn

_ACTIVATIONS = {
    "tanh": nn.Tanh,
    "relu": nn.ReLU,
    "sigmoid": nn.Sigmoid,
    "elu": nn.ELU,
    "linear": nn.Identity,
    None: nn.Identity,
}


def activation_module(name: Optional[str]) -> nn.Module:
    if name not in _ACTIVATIONS:
        raise ValueError(f"Unsupported activation {name!r}; valid: {sorted(k for k in _ACTIVATIONS if k)}")
    return _ACTIVATIONS[name]()


class FeedForwardNet(nn.Module):
    """Dense encoder/decoder stack: n_features → dims... → n_features_out.

    ``dims`` excludes the input width; ``funcs[i]`` is the activation after
    layer i; the final layer uses ``out_func``.
    """

    def __init__(
        self,
        n_features: int,
        n_features_out: int,
        dims: Sequence[int],
        funcs: Sequence[Optional[str]],
        out_func: Optional[str] = "linear",
    ):
        super().__init__()
        if len(dims) != len(funcs):
            raise ValueError("dims and funcs must have equal length")
        self.n_features = n_features
        self.n_features_out = n_features_out
        self.dims: List[int] = list(dims)
        self.funcs: List[Optional[str]] = list(funcs)
        self.out_func = out_func

        layers: List[nn.Module] = []
        widths = [n_features] + self.dims + [n_features_out]
        acts = self.funcs + [out_func]
        for i in range(len(widths) - 1):
            layers.append(nn.Linear(widths[i], widths[i + 1]))
            layers.append(activation_module(acts[i]))
        self.net = nn.Sequential(*layers)

    @property
    def layer_widths(self) -> List[int]:
        return [self.n_features] + self.dims + [self.n_features_out]

    @property
    def layer_funcs(self) -> List[Optional[str]]:
        return self.funcs + [self.out_func]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class LSTMNet(nn.Module):
    """Stacked-LSTM sequence model: window [B, lookback, n_features] →
    prediction [B, n_features_out] (the reference's LSTM autoencoder/forecast
    head: LSTM stack, last timestep's hidden state, Dense out).
    """

    def __init__(
        self,
        n_features: int,
        n_features_out: int,
        dims: Sequence[int],
        funcs: Sequence[Optional[str]],
        out_func: Optional[str] = "linear",
        lookback_window: int = 144,
    ):
        super().__init__()
        if len(dims) != len(funcs):
            raise ValueError("dims and funcs must have equal length")
        self.n_features = n_features
        self.n_features_out = n_features_out
        self.dims = list(dims)
        self.funcs = list(funcs)
        self.out_func = out_func
        self.lookback_window = lookback_window

        cells: List[nn.Module] = []
        widths = [n_features] + self.dims
        for i in range(len(self.dims)):
            cells.append(nn.LSTM(widths[i], widths[i + 1], batch_first=True))
        self.cells = nn.ModuleList(cells)
        self.head = nn.Linear(widths[-1], n_features_out)
        self.head_act = activation_module(out_func)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = x
        for cell in self.cells:
            h, _ = cell(h)
        last = h[:, -1, :]
        return self.head_act(self.head(last))
