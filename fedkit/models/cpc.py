"""Contrastive Predictive Coding trio for 8-channel 32x32 radio patches.

Architecture parity with reference src/simple_models.py:436-514:
  EncoderCNN    (436-470): multi-dilation 4x4 conv bank d in {1,2,4,8,16}
                           concatenated, then 3 stride-2 convs + avg-pool;
                           10,656,552 params at latent_dim=1024
  ContextgenCNN (474-494): pixelCNN-ish 1x1/2x2 conv context generator
  PredictorCNN  (498-514): two 1x1 convs mapping latents/context -> reduced dim
"""

import torch
import torch.nn as nn

from ..ops.conv import FedConvGeneric
from ..ops.pool import avg_pool2d
import torch.nn.functional as F

from ..ops.elu import elu


class EncoderCNN(nn.Module):
    def __init__(self, latent_dim=1024):
        super().__init__()
        self.latent_dim = latent_dim
        # dilated 4x4 stride-2 bank, all 8->8 channels, 32x32 -> 16x16
        self.conv1_1 = FedConvGeneric(8, 8, 4, stride=2, dilation=1, padding=1)
        self.conv1_2 = FedConvGeneric(8, 8, 4, stride=2, dilation=2, padding=3)
        self.conv1_4 = FedConvGeneric(8, 8, 4, stride=2, dilation=4, padding=6)
        self.conv1_8 = FedConvGeneric(8, 8, 4, stride=2, dilation=8, padding=12)
        self.conv1_16 = FedConvGeneric(8, 8, 4, stride=2, dilation=16, padding=24)
        self.conv2 = FedConvGeneric(8 * 5, latent_dim // 4, 4, stride=2, padding=1)   # 16 -> 8
        self.conv3 = FedConvGeneric(latent_dim // 4, latent_dim // 2, 4, stride=2, padding=1)  # 8 -> 4
        self.conv4 = FedConvGeneric(latent_dim // 2, latent_dim, 4, stride=2, padding=1)       # 4 -> 2

    def forward(self, x):
        from ..ops import native_enabled
        from ..ops.conv import dilated_bank
        mods = (self.conv1_1, self.conv1_2, self.conv1_4, self.conv1_8,
                self.conv1_16)
        if native_enabled(x):
            # fused 5-tap bank: ONE MFMA launch for all dilations
            # (block-diagonal combined weight); elu(conv_t(x)+b_t) cat ==
            # elu(bank + b) since elu is elementwise and channels disjoint
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if x.dtype == torch.bfloat16:
                bank = elu(dilated_bank(x, mods))
            else:
                bank = torch.cat([elu(m(x)) for m in mods], dim=1)
        else:
            bank = torch.cat([elu(m(x)) for m in mods], dim=1)
        h = elu(self.conv2(bank))
        h = elu(self.conv3(h))
        h = elu(self.conv4(h))
        return avg_pool2d(h, 2).squeeze()

    def train_order_block_ids(self):
        return [[0, 9], [10, 15]]

    def linear_layer_ids(self):
        return []


class ContextgenCNN(nn.Module):
    def __init__(self, latent_dim=1024):
        super().__init__()
        self.latent_dim = latent_dim
        self.conv1 = FedConvGeneric(latent_dim, latent_dim // 4, 1, stride=1, padding=0, bias=False)
        self.conv2 = FedConvGeneric(latent_dim // 4, latent_dim // 4, 2, stride=1, padding=1, bias=False)
        self.conv3 = FedConvGeneric(latent_dim // 4, latent_dim // 2, 2, stride=1, padding=0, bias=False)
        self.conv4 = FedConvGeneric(latent_dim // 2, latent_dim, 1, stride=1, padding=0, bias=False)

    def forward(self, x):
        x = elu(self.conv1(x))
        x = elu(self.conv2(x))
        x = elu(self.conv3(x))
        return elu(self.conv4(x))

    def train_order_block_ids(self):
        return [[0, 3]]

    def linear_layer_ids(self):
        return []


class PredictorCNN(nn.Module):
    def __init__(self, latent_dim=1024, reduced_dim=64):
        super().__init__()
        self.latent_dim = latent_dim
        self.reduced_dim = reduced_dim
        self.conv1 = FedConvGeneric(latent_dim, reduced_dim, 1, bias=False)
        self.conv2 = FedConvGeneric(latent_dim, reduced_dim, 1, bias=False)

    def forward(self, latents, context):
        return self.conv1(latents), self.conv2(context)

    def train_order_block_ids(self):
        return [[0, 1]]

    def linear_layer_ids(self):
        return []
