"""Model -> kernel-plan descriptors for the stacked HIP engine.

The fast path stores every node replica's parameters as one row of a flat
``[L, n]`` stack (same element order as
``torch.nn.utils.parameters_to_vector``). A :class:`ModelSpec` describes the
sequence of fused kernels that implement the model's forward/backward over
that row, with byte offsets into the flat vector — so the engine never walks
``nn.Module`` objects in the hot loop.

Supported layer kinds:
  ``conv_pool``  Conv2d(1, F, k) + ReLU + MaxPool2d(2)  (MNISTConvNet head)
  ``linear``     Linear with a fused activation: none | relu | sin_relu |
                 sigmoid | tanh | logsoftmax  (``sin_relu`` is the
                 reference FourierNet's relu(sin(scale*Wx+b)) encode)
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

import torch

from .fourier import FourierNet
from .mlp import FFReLUNet, FFSigmoidNet, FFTanhNet
from .mnist_conv import MNISTConvNet


@dataclass
class LayerSpec:
    kind: str                 # 'conv_pool' | 'linear'
    in_dim: int               # linear: in features; conv: image side (28)
    out_dim: int              # linear: out features; conv: num filters
    activation: str           # none|relu|sin_relu|sigmoid|tanh|logsoftmax
    w_off: int                # element offset of the weight in the flat vec
    b_off: int                # element offset of the bias
    scale: float = 1.0        # sin activation scale
    kernel_size: int = 0      # conv only
    # dims of the activation tensor this layer OUTPUTS per sample
    out_elems: int = 0


@dataclass
class ModelSpec:
    name: str
    n: int                    # total flat parameter count per replica
    layers: List[LayerSpec] = field(default_factory=list)
    in_elems: int = 0         # input elements per sample

    @property
    def max_act_elems(self) -> int:
        return max(
            [self.in_elems] + [l.out_elems for l in self.layers]
        )


def param_layout(model: torch.nn.Module):
    """(name, shape, offset) for each parameter, in parameters_to_vector
    order over the flat vector."""
    out = []
    off = 0
    for name, p in model.named_parameters():
        out.append((name, tuple(p.shape), off))
        off += p.numel()
    return out, off


def model_spec(model: torch.nn.Module) -> ModelSpec:
    """Derive the kernel plan for a supported model instance."""
    layout, n = param_layout(model)
    offs = {name: off for name, _, off in layout}

    if isinstance(model, MNISTConvNet):
        F, k, W = model.num_filters, model.kernel_size, model.linear_width
        pool_out = model.pool_out
        spec = ModelSpec(name="mnist_conv", n=n, in_elems=28 * 28)
        spec.layers = [
            LayerSpec(
                kind="conv_pool", in_dim=28, out_dim=F, activation="relu",
                w_off=offs["conv.weight"], b_off=offs["conv.bias"],
                kernel_size=k, out_elems=F * pool_out * pool_out,
            ),
            LayerSpec(
                kind="linear", in_dim=model.fc1_indim, out_dim=W,
                activation="relu", w_off=offs["fc1.weight"],
                b_off=offs["fc1.bias"], out_elems=W,
            ),
            LayerSpec(
                kind="linear", in_dim=W, out_dim=10,
                activation="logsoftmax", w_off=offs["fc2.weight"],
                b_off=offs["fc2.bias"], out_elems=10,
            ),
        ]
        return spec

    if isinstance(model, FourierNet):
        shape = model.shape
        spec = ModelSpec(name="fourier", n=n, in_elems=shape[0])
        spec.layers.append(
            LayerSpec(
                kind="linear", in_dim=shape[0], out_dim=shape[1],
                activation="sin_relu", scale=model.scale,
                w_off=offs["encode.linear.weight"],
                b_off=offs["encode.linear.bias"], out_elems=shape[1],
            )
        )
        nh = len(model.hidden)
        for i in range(nh):
            act = "sigmoid" if i == nh - 1 else "relu"
            spec.layers.append(
                LayerSpec(
                    kind="linear", in_dim=shape[i + 1], out_dim=shape[i + 2],
                    activation=act,
                    w_off=offs[f"hidden.{i}.weight"],
                    b_off=offs[f"hidden.{i}.bias"],
                    out_elems=shape[i + 2],
                )
            )
        return spec

    if isinstance(model, (FFReLUNet, FFTanhNet, FFSigmoidNet)):
        shape = model.shape
        spec = ModelSpec(name=type(model).__name__, n=n, in_elems=shape[0])
        nl = len(model.layers)
        for i in range(nl):
            if i == nl - 1:
                act = model.act_name if model.activate_last else "none"
            else:
                act = model.act_name
            spec.layers.append(
                LayerSpec(
                    kind="linear", in_dim=shape[i], out_dim=shape[i + 1],
                    activation=act,
                    w_off=offs[f"layers.{i}.weight"],
                    b_off=offs[f"layers.{i}.bias"],
                    out_elems=shape[i + 1],
                )
            )
        return spec

    raise NotImplementedError(
        f"No stacked-engine spec for model type {type(model).__name__}"
    )
