"""Implicit-density networks (SIREN-first-layer MLP).

Same architecture as the reference's ``models/fourier_nn.py:14-62``:
a sine-activated first layer f(x) = sin(scale * (W x + b)) with SIREN
init U(-sqrt(6/out), sqrt(6/out)) on the weight, followed by ReLU linear
layers and a Sigmoid output. Unlike the reference, this module does NOT
flip the global default dtype to float64 on import; precision is an
engine/config knob.
"""

import math

import torch
from torch import nn


class SIRENLayer(nn.Module):
    """Sine-activated linear layer: f(x) = sin(scale * linear(x))."""

    def __init__(self, in_features: int, out_features: int, scale: float = 1.0):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.scale = scale
        self.linear = nn.Linear(in_features, out_features)
        c = math.sqrt(6.0 / out_features)
        with torch.no_grad():
            self.linear.weight.uniform_(-c, c)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.sin(self.scale * self.linear(x))


class FourierNet(nn.Module):
    """SIREN encode -> ReLU MLP -> Sigmoid head.

    ``shape`` lists layer widths including input and output, e.g. the paper
    config [2, 256, 64, 64, 64, 1] (n = 25,601 params with scale 0.05).
    """

    def __init__(self, shape, scale: float = 1.0):
        super().__init__()
        self.shape = list(shape)
        self.scale = scale
        self.encode = SIRENLayer(shape[0], shape[1], scale=scale)
        hidden = []
        for i in range(1, len(shape) - 1):
            hidden.append(nn.Linear(shape[i], shape[i + 1]))
        self.hidden = nn.ModuleList(hidden)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # NB: the reference (models/fourier_nn.py:44-58) inserts a ReLU after
        # every non-final layer INCLUDING the SIREN encode, so the first
        # feature map is relu(sin(scale * Wx + b)). Kept for parity.
        x = torch.relu(self.encode(x))
        last = len(self.hidden) - 1
        for i, layer in enumerate(self.hidden):
            x = layer(x)
            x = torch.sigmoid(x) if i == last else torch.relu(x)
        return x
