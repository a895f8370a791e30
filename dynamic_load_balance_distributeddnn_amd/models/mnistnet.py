"""Small MNIST-shape conv net (parity with reference Net/MnistNet.py).

conv5x5(1->10) -> maxpool2 -> ReLU -> conv5x5(10->20) -> Dropout2d ->
maxpool2 -> ReLU -> fc(320->50) -> ReLU -> dropout -> fc(50->10) ->
log_softmax.  21,840 parameters.

Quirk preserved deliberately: the reference trains this with
F.cross_entropy ON TOP of the model's log_softmax output (dbs.py:374 +
Net/MnistNet.py:27) — a double log-softmax.  Mathematically trainable and
kept for behavioral parity.
"""

from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as FD
from ..ops.layers import Conv2d, Linear


class MnistNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = Conv2d(1, 10, 5, bias=True)
        self.conv2 = Conv2d(10, 20, 5, bias=True)
        self.drop2d = nn.Dropout2d()
        self.fc1 = Linear(320, 50)
        self.fc2 = Linear(50, 10)

    def forward(self, x):
        x = F.relu(FD.max_pool2d(self.conv1(x), 2))
        x = F.relu(FD.max_pool2d(self.drop2d(self.conv2(x)), 2))
        x = x.flatten(1)
        x = F.dropout(F.relu(self.fc1(x)), training=self.training)
        return FD.log_softmax(self.fc2(x), dim=-1)
