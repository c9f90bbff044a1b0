"""Model zoo with the reference's architectures and block partitions.

Every model exposes
  train_order_block_ids() -> list of [lo, hi] parameter-tensor index ranges
      (the hand-written training-block partition the federation cycles over;
       reference: src/simple_models.py:38-39, 222-226, 304-305, 430-432, ...)
  linear_layer_ids() -> list of parameter-tensor indices of fc layers
      (used to gate L1/L2 regularization; reference: simple_models.py:29-30)
"""

from .classifiers import Net, Net1, Net2
from .resnet import BasicBlock, Bottleneck, ResNet, ResNet18, ResNet9
from .vae import AutoEncoderCNN, AutoEncoderCNNCL
from .cpc import EncoderCNN, ContextgenCNN, PredictorCNN

MODEL_FACTORIES = {
    "Net": Net,
    "Net1": Net1,
    "Net2": Net2,
    "ResNet18": ResNet18,
    "ResNet9": ResNet9,
    "AutoEncoderCNN": AutoEncoderCNN,
    "AutoEncoderCNNCL": AutoEncoderCNNCL,
}

__all__ = [
    "Net", "Net1", "Net2",
    "BasicBlock", "Bottleneck", "ResNet", "ResNet18", "ResNet9",
    "AutoEncoderCNN", "AutoEncoderCNNCL",
    "EncoderCNN", "ContextgenCNN", "PredictorCNN",
    "MODEL_FACTORIES",
]
