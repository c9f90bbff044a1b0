"""Loss parity tests: the vectorized losses must equal the reference's
Python-loop formulations on the same inputs."""

import math

import torch
import torch.nn.functional as F

from fedkit.ops import losses as L


def test_cross_entropy_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(32, 10)
    labels = torch.randint(0, 10, (32,))
    assert torch.allclose(L.cross_entropy(logits, labels),
                          F.cross_entropy(logits, labels))


def test_vae_loss():
    torch.manual_seed(0)
    x = torch.rand(4, 3, 8, 8)
    rx = torch.rand(4, 3, 8, 8)
    mu = torch.randn(4, 10)
    logvar = torch.randn(4, 10)
    got = L.vae_loss(rx, x, mu, logvar)
    want = F.mse_loss(rx, x, reduction="sum") \
        - 0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp())
    assert torch.allclose(got, want)


def _ref_cost1(pk, mu, sig2, x):
    """Reference federated_vae_cl.py:101-109 verbatim semantics (batch loop)."""
    b = x.shape[0]
    err = (x - mu).pow(2).div(2 * sig2)
    err1 = 0.5 * torch.log(sig2 * 2 * math.pi)
    loss = 0
    for ci in range(b):
        loss = loss + pk[ci] * torch.sum(err[ci] + err1[ci])
    return loss / b


def _ref_cost3(pk, q_mu, q_sig2, p_mu, p_sig2):
    b = pk.shape[0]
    mudiff = (p_mu - q_mu).pow(2).div(p_sig2)
    sigratio = q_sig2 / p_sig2
    loss = 0
    for ci in range(b):
        loss = loss + 0.5 * pk[ci] * torch.sum(
            sigratio[ci] - torch.log(sigratio[ci]) + mudiff[ci] - 1)
    return loss / b


def test_vaecl_costs_match_reference_loops():
    torch.manual_seed(1)
    B = 6
    pk = torch.rand(B)
    x = torch.rand(B, 3, 8, 8)
    mu = torch.rand(B, 3, 8, 8)
    sig2 = torch.rand(B, 3, 8, 8) + 0.1
    assert torch.allclose(L.cost1(pk, mu, sig2, x), _ref_cost1(pk, mu, sig2, x),
                          atol=1e-5)
    q_mu, p_mu = torch.randn(B, 16), torch.randn(B, 16)
    q_s2 = torch.rand(B, 16) + 0.1
    p_s2 = torch.rand(B, 16) + 0.1
    assert torch.allclose(L.cost3(pk, q_mu, q_s2, p_mu, p_s2),
                          _ref_cost3(pk, q_mu, q_s2, p_mu, p_s2), atol=1e-5)
    # cost2 / cost21 closed forms
    want2 = -(pk * torch.log(pk + 1e-9)).sum() / B
    assert torch.allclose(L.cost2(pk), want2)
    pbar = pk.mean()
    want21 = 1 / (-pbar * torch.log(pbar + 1e-9) + 1e-9)
    assert torch.allclose(L.cost21(pk), want21)


def _ref_info_nce(z, zhat):
    """Reference federated_cpc.py:149-180 O(p^4) loop, verbatim semantics."""
    (nbatch, nchan, px, py) = z.shape
    Z = z.reshape(-1, px * py)
    Zhat = zhat.reshape(-1, px * py)
    P = px * py
    zz = torch.zeros(P, P)
    for ci in range(P):
        znrm = torch.norm(Z[:, ci])
        for cj in range(P):
            zz[ci, cj] = torch.dot(Z[:, ci], Zhat[:, cj]) / (znrm * torch.norm(Zhat[:, cj]))
    loss = 0
    for ci in range(P):
        numerator = torch.exp(zz[ci, ci])
        denominator = numerator
        for cj in [i for i in range(P) if i != ci]:
            denominator = denominator + torch.exp(zz[ci, cj])
        loss = loss - torch.log(numerator / denominator + 1e-6)
    return loss


def test_info_nce_matches_reference_loop():
    torch.manual_seed(2)
    z = torch.randn(4, 8, 3, 3)
    zhat = torch.randn(4, 8, 3, 3)
    got = L.info_nce(z, zhat)
    want = _ref_info_nce(z, zhat)
    assert torch.allclose(got, want, atol=1e-4)


def test_info_nce_gradients_flow():
    z = torch.randn(2, 4, 2, 2, requires_grad=True)
    zhat = torch.randn(2, 4, 2, 2, requires_grad=True)
    L.info_nce(z, zhat).backward()
    assert z.grad is not None and torch.isfinite(z.grad).all()
    assert zhat.grad is not None and torch.isfinite(zhat.grad).all()
