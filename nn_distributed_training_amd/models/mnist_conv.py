"""MNIST convolutional classifier.

Same architecture as the reference's ``models/mnist_conv_nn.py:4-28``:
Conv2d(1, F, k) -> ReLU -> MaxPool2d(2) -> Flatten -> Linear -> ReLU ->
Linear(W, 10) -> LogSoftmax.  Paper config (F=3, k=5, W=64) has n=28,440
parameters.

On the HIP fast path this module is *not* executed by torch: the stacked
engine (ops/stacked.py) runs the equivalent fused CDNA4 kernels over all
node replicas at once; this class defines the architecture, the parameter
layout (via spec.model_spec) and the eager/golden execution path.
"""

import torch
from torch import nn


class MNISTConvNet(nn.Module):
    def __init__(self, num_filters: int, kernel_size: int, linear_width: int):
        super().__init__()
        self.num_filters = num_filters
        self.kernel_size = kernel_size
        self.linear_width = linear_width
        conv_out = 28 - (kernel_size - 1)
        self.pool_out = conv_out // 2
        self.fc1_indim = num_filters * self.pool_out**2

        self.conv = nn.Conv2d(1, num_filters, kernel_size, 1)
        self.fc1 = nn.Linear(self.fc1_indim, linear_width)
        self.fc2 = nn.Linear(linear_width, 10)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = torch.relu(self.conv(x))
        x = torch.nn.functional.max_pool2d(x, 2)
        x = torch.flatten(x, 1)
        x = torch.relu(self.fc1(x))
        return torch.log_softmax(self.fc2(x), dim=1)
