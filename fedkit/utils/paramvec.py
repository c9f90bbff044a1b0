"""Parameter-vector / freeze utilities — the federation ABI.

Behavior parity with reference src/simple_utils.py:9-87; the flatten /
scatter pair (get_trainable_values / put_trainable_values) is the entire
federation data plane.  On MI355X the flatten and scatter run
as ONE multi-tensor HIP kernel (fedkit.ops.flat) instead of the reference's
per-tensor Python loop of copies.
"""

from typing import List, Optional

import torch
import torch.nn as nn

from ..ops import flat as flat_ops


def init_weights(m: nn.Module) -> None:
    """Xavier-uniform weight init, bias = 0.01 (simple_utils.py:9-14).

    Applied under a fixed torch.manual_seed by the drivers so every client
    starts identical; in the distributed engine every rank runs this with the
    same seed instead of broadcasting (same result, zero traffic).
    """
    if isinstance(m, (nn.Linear, nn.Conv2d)):
        torch.nn.init.xavier_uniform_(m.weight)
        if getattr(m, "bias", None) is not None:
            m.bias.data.fill_(0.01)


def unfreeze_one_layer(net: nn.Module, layer_id: int) -> None:
    """Only layer `layer_id` trainable: parameter tensors 2i and 2i+1
    (simple_utils.py:16-22)."""
    for ci, p in enumerate(net.parameters()):
        p.requires_grad = ci in (2 * layer_id, 2 * layer_id + 1)


def unfreeze_all_layers(net: nn.Module) -> None:
    for p in net.parameters():
        p.requires_grad = True


def freeze_all_layers(net: nn.Module) -> None:
    for p in net.parameters():
        p.requires_grad = False


def unfreeze_one_block(net: nn.Module, blockid: int) -> None:
    """Only parameter tensors [lo..hi] of block `blockid` trainable
    (simple_utils.py:34-45).

    Accepts either a block index (reference semantics) or directly a [lo,hi]
    pair — the reference's federated_cpc.py:219-223 passes the pair where an
    index is expected and crashes (SURVEY.md C14 latent bug); supporting both
    fixes CPC without changing the other drivers.
    """
    if isinstance(blockid, (list, tuple)):
        lo, hi = blockid
    else:
        lo, hi = net.train_order_block_ids()[blockid]
    for ci, p in enumerate(net.parameters()):
        p.requires_grad = lo <= ci <= hi


def trainable_params(net: nn.Module) -> List[torch.Tensor]:
    return [p for p in net.parameters() if p.requires_grad]


def flat_physical(p: torch.Tensor) -> torch.Tensor:
    """Differentiable flat view of p in PHYSICAL (storage) element order —
    the order the fused pack/unpack/axpy kernels traverse.  Penalty terms
    built from parameters (FedProx/ADMM closures) must use this so
    (x - z) pairs elements consistently with the packed federation
    vectors; channels_last conv weights differ from .reshape(-1)'s
    logical order.  Zero-copy for standard and channels_last layouts."""
    if p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last):
        return p.permute(0, 2, 3, 1).reshape(-1)
    return p.reshape(-1)


def flat_trainable(net: nn.Module) -> torch.Tensor:
    """Differentiable physical-order flat vector of the trainable params
    (the closure-side counterpart of get_trainable_values)."""
    return torch.cat([flat_physical(p) for p in trainable_params(net)])


def get_trainable_values(net: nn.Module, mydevice=None) -> torch.Tensor:
    """Flatten trainable parameters into one contiguous fp32 vector
    (simple_utils.py:47-66) — one fused kernel on GPU."""
    params = trainable_params(net)
    with torch.no_grad():
        vec = flat_ops.pack([p.detach() for p in params])
    if mydevice is not None and vec.device != torch.device(mydevice):
        vec = vec.to(mydevice)
    return vec


def put_trainable_values(net: nn.Module, x: torch.Tensor) -> None:
    """Scatter a flat vector back into the trainable parameters
    (simple_utils.py:68-77) — one fused kernel on GPU."""
    params = trainable_params(net)
    with torch.no_grad():
        flat_ops.unpack(x, [p.detach() for p in params])


def number_of_layers(net: nn.Module) -> int:
    """Total number of parameter TENSORS (not /2; simple_utils.py:79-83)."""
    return sum(1 for _ in net.parameters())


def number_of_blocks(net: nn.Module) -> int:
    return len(net.train_order_block_ids())
