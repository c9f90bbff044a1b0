from .paramvec import (
    init_weights,
    freeze_all_layers,
    unfreeze_all_layers,
    unfreeze_one_layer,
    unfreeze_one_block,
    flat_physical, flat_trainable, get_trainable_values,
    put_trainable_values,
    number_of_layers,
    number_of_blocks,
    trainable_params,
)
from .checkpoint import save_client_checkpoint, load_client_checkpoint

__all__ = [
    "init_weights", "freeze_all_layers", "unfreeze_all_layers",
    "unfreeze_one_layer", "unfreeze_one_block",
    "get_trainable_values", "put_trainable_values",
    "number_of_layers", "number_of_blocks", "trainable_params",
    "save_client_checkpoint", "load_client_checkpoint",
]
