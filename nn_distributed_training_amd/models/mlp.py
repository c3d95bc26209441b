"""Generic feed-forward MLP families.

Parity with the reference's ``models/relu_nn.py``:
  - FFReLUNet  (:4-40):  ReLU between layers, linear output
  - FFTanhNet  (:43-78): Tanh after every layer including the last
  - FFSigmoidNet (:81-116): Sigmoid after every layer including the last
All take ``shape`` = list of widths including input and output.
"""

import torch
from torch import nn


class _FFNet(nn.Module):
    #: activation applied between layers; subclass sets these
    act = staticmethod(torch.relu)
    act_name = "relu"
    #: whether the activation is also applied after the final layer
    activate_last = False

    def __init__(self, shape):
        super().__init__()
        self.shape = list(shape)
        self.layers = nn.ModuleList(
            nn.Linear(shape[i], shape[i + 1]) for i in range(len(shape) - 1)
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        last = len(self.layers) - 1
        for i, layer in enumerate(self.layers):
            x = layer(x)
            if i != last or self.activate_last:
                x = self.act(x)
        return x


class FFReLUNet(_FFNet):
    act = staticmethod(torch.relu)
    act_name = "relu"
    activate_last = False


class FFTanhNet(_FFNet):
    act = staticmethod(torch.tanh)
    act_name = "tanh"
    activate_last = True


class FFSigmoidNet(_FFNet):
    act = staticmethod(torch.sigmoid)
    act_name = "sigmoid"
    activate_last = True
