"""Plain CNN classifiers for CIFAR10 (architecture parity with the reference).

Architectures, parameter counts and block partitions match
reference src/simple_models.py:
  Net   -> simple_models.py:9-39   (LeNet-style, 62,006 params / 10 tensors)
  Net1  -> simple_models.py:42-77  (VGG-ish,     890,410 params / 12 tensors)
  Net2  -> simple_models.py:81-128 (deeper VGG-ish, 2,513,418 params / 18 tensors)
All activations are ELU (the reference replaced ReLU everywhere).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.conv import FedConvGeneric
from ..ops.linear import FedLinear
from ..ops.elu import elu
from ..ops.pool import FedMaxPool2d


class Net(nn.Module):
    """LeNet-style 5x5-conv CIFAR10 classifier (simple_models.py:9-39)."""

    def __init__(self):
        super().__init__()
        self.conv1 = FedConvGeneric(3, 6, 5)
        self.pool = FedMaxPool2d(2, 2)
        self.conv2 = FedConvGeneric(6, 16, 5)
        self.fc1 = FedLinear(16 * 5 * 5, 120)
        self.fc2 = FedLinear(120, 84)
        self.fc3 = FedLinear(84, 10)

    def forward(self, x):
        x = self.pool(elu(self.conv1(x)))
        x = self.pool(elu(self.conv2(x)))
        x = x.reshape(-1, 16 * 5 * 5)
        x = elu(self.fc1(x))
        x = elu(self.fc2(x))
        return self.fc3(x)

    def linear_layer_ids(self):
        # fc1/fc2/fc3 weight-tensor indices (simple_models.py:29-30).
        # NOTE (reference quirk, kept): drivers compare a BLOCK index against
        # these PARAMETER indices when gating regularization.
        return [4, 6, 8]

    def linear_layer_parameters(self):
        # Reference quirk kept: `a or b` short-circuits, so only fc1's
        # parameters are returned (simple_models.py:33-35); dead code in
        # practice (never called by drivers).
        gen = self.fc1.parameters() or self.fc2.parameters() or self.fc3.parameters()
        return torch.cat([p.view(-1) for p in gen])

    def train_order_block_ids(self):
        return [[4, 5], [0, 1], [2, 3], [6, 7], [8, 9]]


class Net1(nn.Module):
    """4-conv VGG-ish CIFAR10 classifier (simple_models.py:42-77)."""

    def __init__(self):
        super().__init__()
        self.conv1 = FedConvGeneric(3, 32, 3)
        self.conv2 = FedConvGeneric(32, 32, 3)
        self.conv3 = FedConvGeneric(32, 64, 3)
        self.conv4 = FedConvGeneric(64, 64, 3)
        self.pool1 = FedMaxPool2d(2, 2)
        self.pool2 = FedMaxPool2d(2, 2)
        self.fc1 = FedLinear(64 * 5 * 5, 512)
        self.fc2 = FedLinear(512, 10)

    def forward(self, x):
        x = elu(self.conv1(x))            # 32x32 -> 30x30
        x = self.pool1(elu(self.conv2(x)))  # 28x28 -> 14x14
        x = elu(self.conv3(x))            # -> 12x12
        x = self.pool2(elu(self.conv4(x)))  # 10x10 -> 5x5
        x = x.reshape(-1, 64 * 5 * 5)
        x = elu(self.fc1(x))
        return self.fc2(x)

    def linear_layer_ids(self):
        return [8, 10]

    def linear_layer_parameters(self):
        gen = self.fc1.parameters() or self.fc2.parameters()
        return torch.cat([p.view(-1) for p in gen])

    def train_order_block_ids(self):
        return [[4, 5], [10, 11], [2, 3], [6, 7], [0, 1], [8, 9]]


class Net2(nn.Module):
    """4-conv / 5-fc padded VGG-ish CIFAR10 classifier (simple_models.py:81-128)."""

    def __init__(self):
        super().__init__()
        self.conv1 = FedConvGeneric(3, 64, 3, padding=1)
        self.conv2 = FedConvGeneric(64, 128, 3, padding=1)
        self.conv3 = FedConvGeneric(128, 256, 3, padding=1)
        self.conv4 = FedConvGeneric(256, 512, 3, padding=1)
        self.pool1 = FedMaxPool2d(2, 2)
        self.pool2 = FedMaxPool2d(2, 2)
        self.pool3 = FedMaxPool2d(2, 2)
        self.pool4 = FedMaxPool2d(2, 2)
        self.fc1 = FedLinear(512 * 2 * 2, 128)
        self.fc2 = FedLinear(128, 256)
        self.fc3 = FedLinear(256, 512)
        self.fc4 = FedLinear(512, 1024)
        self.fc5 = FedLinear(1024, 10)

    def forward(self, x):
        x = self.pool1(elu(self.conv1(x)))  # 32 -> 16
        x = self.pool2(elu(self.conv2(x)))  # 16 -> 8
        x = self.pool3(elu(self.conv3(x)))  # 8 -> 4
        x = self.pool4(elu(self.conv4(x)))  # 4 -> 2
        x = x.reshape(-1, 512 * 2 * 2)
        x = elu(self.fc1(x))
        x = elu(self.fc2(x))
        x = elu(self.fc3(x))
        x = elu(self.fc4(x))
        return self.fc5(x)

    def linear_layer_ids(self):
        return [12, 14, 16]

    def linear_layer_parameters(self):
        gen = (self.fc1.parameters() or self.fc2.parameters()
               or self.fc3.parameters() or self.fc4.parameters()
               or self.fc5.parameters())
        return torch.cat([p.view(-1) for p in gen])

    def train_order_block_ids(self):
        return [[14, 15], [4, 5], [2, 3], [8, 9], [16, 17],
                [12, 13], [6, 7], [0, 1], [10, 11]]
