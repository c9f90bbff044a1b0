"""Model-zoo parity tests: parameter counts, tensor counts and block
partitions must match the reference's documented values."""

import pytest
import torch

from fedkit import models as M

# (factory, params, tensors, n_blocks) — values from SURVEY.md §2
ZOO = [
    (M.Net, 62006, 10, 5),
    (M.Net1, 890410, 12, 6),
    (M.Net2, 2513418, 18, 9),
    (M.ResNet18, 11173962, 62, 10),
    (M.ResNet9, 4903242, 38, 8),
    (M.AutoEncoderCNN, 205679, 24, 12),
    (M.AutoEncoderCNNCL, 350744, 42, 3),
]


@pytest.mark.parametrize("factory,n_params,n_tensors,n_blocks", ZOO)
def test_param_counts(factory, n_params, n_tensors, n_blocks):
    m = factory()
    assert sum(p.numel() for p in m.parameters()) == n_params
    assert sum(1 for _ in m.parameters()) == n_tensors
    assert len(m.train_order_block_ids()) == n_blocks


@pytest.mark.parametrize("factory,n_params,n_tensors,n_blocks", ZOO)
def test_block_partition_covers_all_tensors(factory, n_params, n_tensors, n_blocks):
    """Every parameter tensor belongs to exactly one training block
    (except the known coarse VAE-CL partition which is also a full cover)."""
    m = factory()
    blocks = m.train_order_block_ids()
    covered = sorted(i for lo, hi in blocks for i in range(lo, hi + 1))
    assert covered == list(range(n_tensors))


@pytest.mark.parametrize("factory", [M.Net, M.Net1, M.Net2, M.ResNet18, M.ResNet9])
def test_classifier_forward_shape(factory):
    m = factory()
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_vae_forward():
    m = M.AutoEncoderCNN()
    out, mu, logvar = m(torch.randn(3, 3, 32, 32))
    assert out.shape == (3, 3, 32, 32)
    assert mu.shape == (3, 10) and logvar.shape == (3, 10)
    assert out.min() >= 0 and out.max() <= 1  # sigmoid output


def test_vaecl_forward_trunk_sharing_matches_reference_semantics():
    """The fan-out forward (trunk computed once) must equal running the
    cluster-conditioned encoder from scratch per cluster."""
    torch.manual_seed(0)
    m = M.AutoEncoderCNNCL(K=4, L=8)
    m.eval()
    x = torch.randn(2, 3, 32, 32)
    ekhat, mu_xi, sig2_xi, *_ = m(x)
    assert ekhat.shape == (2, 4)
    assert torch.allclose(ekhat.sum(dim=1), torch.ones(2), atol=1e-5)
    ek1 = torch.zeros(2, 4)
    ek1[:, 1] = 1
    mu_direct, sig2_direct = m.encode(x, ek1)
    assert torch.allclose(mu_direct, mu_xi[1], atol=1e-6)
    assert torch.allclose(sig2_direct, sig2_xi[1], atol=1e-6)


def test_cpc_shapes():
    enc = M.EncoderCNN(latent_dim=64)
    ctx = M.ContextgenCNN(latent_dim=64)
    pred = M.PredictorCNN(latent_dim=64, reduced_dim=16)
    y = torch.randn(8, 8, 32, 32)
    lat = enc(y)
    assert lat.shape == (8, 64)
    lat_grid = lat.view(2, 2, 2, -1).permute(0, 3, 1, 2).contiguous()
    c = ctx(lat_grid)
    assert c.shape == lat_grid.shape
    rl, pr = pred(lat_grid, c)
    assert rl.shape == (2, 16, 2, 2) and pr.shape == (2, 16, 2, 2)


def test_resnet_state_dict_keys_match_reference_layout():
    """Checkpoint compat: module names follow the reference's attribute
    names (conv1/bn1/layer1..4/linear with BasicBlock conv1/bn1/conv2/bn2/
    shortcut)."""
    m = M.ResNet18()
    keys = set(m.state_dict().keys())
    for expected in [
        "conv1.weight", "bn1.weight", "bn1.running_mean",
        "layer1.0.conv1.weight", "layer2.0.shortcut.0.weight",
        "layer4.1.bn2.running_var", "linear.weight", "linear.bias",
    ]:
        assert expected in keys
