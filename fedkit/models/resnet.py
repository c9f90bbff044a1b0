"""CIFAR ResNet with ELU activations — the flagship federated model.

Architecture parity with reference src/simple_models.py:132-237:
  BasicBlock  (132-154): conv3x3-BN-ELU-conv3x3-BN + (1x1-BN shortcut) + ELU
  Bottleneck  (157-182): defined for parity; the factories never use it
  ResNet      (185-230): conv3x3(3->64)-BN-ELU, 4 stages (64/128/256/512,
                         strides 1/2/2/2), avg-pool 4, fc -> 10
  ResNet18(): 11,173,962 params / 62 tensors, 10-block training partition
  ResNet9():   4,903,242 params / 38 tensors,  8-block training partition

MI355X notes: on ROCm the fast path runs this model in channels_last
(NHWC) bf16 so that conv3x3/1x1 dispatch to the fedkit CDNA4 implicit-GEMM
MFMA kernels and BN+ELU to the fused normalization kernels (fedkit.ops).
The module graph (and therefore state_dict / parameter order / block
partition) is identical to the reference's.
"""

import os

import torch.nn as nn

from ..ops.conv import FedConv2d
from ..ops.linear import FedLinear
from ..ops.pool import avg_pool2d
from ..ops.norm import FedBatchNorm2d, bn_elu


# A/B gate for the bn-apply-into-pad fusion (default on)
_PAD_FUSE = 0 if os.environ.get("FEDKIT_NO_PADFUSE") == "1" else 1


def _conv3x3(cin, cout, stride=1):
    return FedConv2d(cin, cout, kernel_size=3, stride=stride, padding=1, bias=False)


def _conv1x1(cin, cout, stride=1):
    return FedConv2d(cin, cout, kernel_size=1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1):
        super().__init__()
        self.conv1 = _conv3x3(in_planes, planes, stride)
        self.bn1 = FedBatchNorm2d(planes)
        self.conv2 = _conv3x3(planes, planes, 1)
        self.bn2 = FedBatchNorm2d(planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                _conv1x1(in_planes, self.expansion * planes, stride),
                FedBatchNorm2d(self.expansion * planes),
            )

    def forward(self, x):
        # bn1's output feeds ONLY conv2 (pad 1): the apply kernel writes
        # the padded image directly (bn_elu pad_out fusion) and conv2
        # skips its pad pass.  bn2's output is padded too when the NEXT
        # block is identity-shortcut (pad_out_next, set by ResNet): its
        # conv1 consumes the marker and its bn2 reads the padded residual
        # at interior coordinates.
        out = bn_elu(self.bn1, self.conv1(x), pad_out=_PAD_FUSE)
        return bn_elu(self.bn2, self.conv2(out), residual=self.shortcut(x),
                      pad_out=_PAD_FUSE and getattr(self, "pad_out_next", 0))


class Bottleneck(nn.Module):
    """Defined for parity with simple_models.py:157-182; unused by factories."""

    expansion = 4

    def __init__(self, in_planes, planes, stride=1):
        super().__init__()
        self.conv1 = _conv1x1(in_planes, planes)
        self.bn1 = FedBatchNorm2d(planes)
        self.conv2 = _conv3x3(planes, planes, stride)
        self.bn2 = FedBatchNorm2d(planes)
        self.conv3 = _conv1x1(planes, self.expansion * planes)
        self.bn3 = FedBatchNorm2d(self.expansion * planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                _conv1x1(in_planes, self.expansion * planes, stride),
                FedBatchNorm2d(self.expansion * planes),
            )

    def forward(self, x):
        out = bn_elu(self.bn1, self.conv1(x))
        out = bn_elu(self.bn2, self.conv2(out))
        return bn_elu(self.bn3, self.conv3(out), residual=self.shortcut(x))


class ResNet(nn.Module):
    def __init__(self, block, num_blocks, qualifier, num_classes=10):
        super().__init__()
        self.qualifier = qualifier  # 9 or 18
        self.in_planes = 64
        self.conv1 = _conv3x3(3, 64, 1)
        self.bn1 = FedBatchNorm2d(64)
        self.layer1 = self._make_layer(block, 64, num_blocks[0], 1)
        self.layer2 = self._make_layer(block, 128, num_blocks[1], 2)
        self.layer3 = self._make_layer(block, 256, num_blocks[2], 2)
        self.layer4 = self._make_layer(block, 512, num_blocks[3], 2)
        self.linear = FedLinear(512 * block.expansion, num_classes)

    def _make_layer(self, block, planes, n, stride):
        layers = []
        for i, s in enumerate([stride] + [1] * (n - 1)):
            blk = block(self.in_planes, planes, s)
            # bn2 -> next block's conv1 pad fusion: legal when the
            # successor is the NEXT block of the SAME stage (identity
            # shortcut: conv1 takes the marker, the residual is read at
            # interior coords); stage-crossing successors have a 1x1
            # shortcut that needs the unpadded image
            blk.pad_out_next = 1 if (i + 1 < n and block is BasicBlock) else 0
            layers.append(blk)
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def _precast(self, x):
        # one fused kernel casts every TRAINABLE conv weight to bf16 for
        # this step's graph (per-conv casts cost a launch each; frozen
        # weights keep their version-cached copies)
        from ..ops import native_enabled
        if not native_enabled(x):
            return
        mods = self.__dict__.get("_fedconvs")
        if mods is None:
            mods = [m for m in self.modules() if isinstance(m, FedConv2d)]
            self.__dict__["_fedconvs"] = mods
        from ..ops.conv import batch_cast_weights
        train_mods = [m for m in mods if m.weight.requires_grad]
        if len(train_mods) > 1:
            batch_cast_weights(train_mods)

    def forward(self, x):
        self._precast(x)
        # the stem's output feeds layer1.0 (identity shortcut): fused pad
        out = bn_elu(self.bn1, self.conv1(x), pad_out=_PAD_FUSE)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = avg_pool2d(out, 4)
        out = out.reshape(out.size(0), -1)
        return self.linear(out)

    def train_order_block_ids(self):
        # Hand-written parameter-tensor partitions (simple_models.py:222-226).
        if self.qualifier == 18:
            return [[0, 2], [3, 8], [9, 14], [15, 23], [24, 29],
                    [30, 38], [39, 44], [45, 53], [54, 59], [60, 61]]
        return [[0, 2], [3, 8], [9, 14], [15, 17],
                [18, 23], [24, 29], [30, 32], [33, 37]]

    def linear_layer_ids(self):
        return []


def ResNet18():
    return ResNet(BasicBlock, [2, 2, 2, 2], qualifier=18)


def ResNet9():
    return ResNet(BasicBlock, [1, 1, 1, 1], qualifier=9)
