"""Federation-ABI tests: flat pack/unpack round trip, freeze semantics
(SURVEY.md §4 implied test plan)."""

import torch

from fedkit import models as M
from fedkit.ops import flat as flat_ops
from fedkit.utils import (freeze_all_layers, get_trainable_values,
                          number_of_blocks, number_of_layers,
                          put_trainable_values, trainable_params,
                          unfreeze_one_block, unfreeze_one_layer)


def test_pack_unpack_roundtrip():
    torch.manual_seed(0)
    tensors = [torch.randn(3, 4), torch.randn(7), torch.randn(2, 2, 2)]
    flat = flat_ops.pack(tensors)
    assert flat.shape == (3 * 4 + 7 + 8,)
    outs = [torch.zeros_like(t) for t in tensors]
    flat_ops.unpack(flat, outs)
    for a, b in zip(tensors, outs):
        assert torch.equal(a, b)


def test_add_flat_matches_manual():
    torch.manual_seed(1)
    tensors = [torch.randn(5), torch.randn(3, 3)]
    ref = [t.clone() for t in tensors]
    upd = torch.randn(14)
    flat_ops.add_flat(tensors, upd, 0.25)
    off = 0
    for t, r in zip(tensors, ref):
        n = r.numel()
        assert torch.allclose(t, r + 0.25 * upd[off:off + n].view_as(r))
        off += n


def test_unfreeze_one_block_resnet18():
    m = M.ResNet18()
    unfreeze_one_block(m, 3)   # block [15,23]
    ids = [i for i, p in enumerate(m.parameters()) if p.requires_grad]
    assert ids == list(range(15, 24))
    # pair form (the CPC driver fix) is accepted too
    unfreeze_one_block(m, [3, 8])
    ids = [i for i, p in enumerate(m.parameters()) if p.requires_grad]
    assert ids == list(range(3, 9))


def test_unfreeze_one_layer():
    m = M.Net()
    unfreeze_one_layer(m, 2)
    ids = [i for i, p in enumerate(m.parameters()) if p.requires_grad]
    assert ids == [4, 5]


def test_get_put_trainable_values_roundtrip():
    torch.manual_seed(2)
    m = M.Net()
    unfreeze_one_block(m, 0)   # fc1: tensors 4,5
    v = get_trainable_values(m)
    assert v.numel() == sum(p.numel() for p in trainable_params(m))
    v2 = torch.randn_like(v)
    put_trainable_values(m, v2)
    assert torch.allclose(get_trainable_values(m), v2)
    # frozen params untouched by put
    frozen_before = [p.clone() for p in m.parameters() if not p.requires_grad]
    put_trainable_values(m, torch.randn_like(v))
    frozen_after = [p for p in m.parameters() if not p.requires_grad]
    for a, b in zip(frozen_before, frozen_after):
        assert torch.equal(a, b)


def test_layer_block_counts():
    m = M.ResNet18()
    assert number_of_layers(m) == 62
    assert number_of_blocks(m) == 10
    freeze_all_layers(m)
    assert len(trainable_params(m)) == 0


def test_cpu_fallback_physical_order_channels_last():
    """The CPU fallback pack/unpack/add_flat must traverse PHYSICAL
    (storage) order for channels_last tensors, matching flat_physical and
    the native kernels."""
    from fedkit.utils.paramvec import flat_physical
    torch.manual_seed(5)
    w = torch.randn(4, 6, 3, 3).contiguous(memory_format=torch.channels_last)
    b = torch.randn(6)
    flat = flat_ops.pack([w, b])
    expect = torch.cat([flat_physical(w), b])
    assert torch.equal(flat, expect)
    # unpack round trip lands values back in the right places
    w2 = torch.empty_like(w).contiguous(memory_format=torch.channels_last)
    b2 = torch.empty_like(b)
    flat_ops.unpack(flat, [w2, b2])
    assert torch.equal(w2, w) and torch.equal(b2, b)
    # add_flat pairs elements physically too
    w3 = w.clone()
    flat_ops.add_flat([w3, b2], flat, 1.0)
    assert torch.allclose(w3, 2 * w)


def test_unpack_bumps_version_counter():
    """Derived-weight caches key on param._version; unpack/add_flat must
    advance it on every write path."""
    t = torch.randn(8)
    v0 = t._version
    flat_ops.unpack(torch.randn(8), [t])
    assert t._version > v0
    v1 = t._version
    flat_ops.add_flat([t], torch.randn(8), 0.5)
    assert t._version > v1
