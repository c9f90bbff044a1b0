from .cifar import (
    shard_indices, make_client_datasets, DeviceShardLoader, CIFAR_TRAIN_N,
    CIFAR_TEST_N, client_normalization,
)
from .lofar import lofar_minibatch, synthetic_visibilities, patch_visibilities

__all__ = [
    "shard_indices", "make_client_datasets", "DeviceShardLoader",
    "CIFAR_TRAIN_N", "CIFAR_TEST_N", "client_normalization",
    "lofar_minibatch", "synthetic_visibilities", "patch_visibilities",
]
