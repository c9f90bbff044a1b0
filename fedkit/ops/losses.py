"""Loss functions: fused/vectorized forms of the reference's losses.

cross_entropy      -> reference nn.CrossEntropyLoss (federated_multi.py:130-132);
                      GPU path is a fused log-softmax+NLL HIP kernel.
vae_loss           -> federated_vae.py:97-108 (MSE(sum) + analytic KLD).
cost1/2/21/3,
vaecl_loss         -> federated_vae_cl.py:101-162.  The reference loops over
                      the batch in Python (cost1/cost2/cost3); here the same
                      sums are single device reductions (identical math,
                      different fp summation order).
info_nce           -> federated_cpc.py:149-180.  The reference builds the
                      (px*py)^2 cosine-similarity matrix with an O(p^4)
                      Python loop of torch.dot; here it is ONE GEMM of the
                      column-normalized matrices + a row softmax (identical
                      values).
"""

import math

import torch
import torch.nn.functional as F


def _native(t):
    from . import native_enabled
    return native_enabled(t)


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        from . import require_ext
        loss, lse = require_ext().cross_entropy_fwd(logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return loss

    @staticmethod
    def backward(ctx, gloss):
        from . import require_ext
        logits, labels, lse = ctx.saved_tensors
        gx = require_ext().cross_entropy_bwd(logits, labels, lse, gloss)
        return gx, None


def cross_entropy(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean-reduced cross entropy (== nn.CrossEntropyLoss()(logits, labels))."""
    if _native(logits) and logits.dim() == 2:
        return _CrossEntropyFn.apply(logits.contiguous(), labels.contiguous())
    return F.cross_entropy(logits, labels)


class CrossEntropyLoss(torch.nn.Module):
    """Drop-in for nn.CrossEntropyLoss() as the drivers construct it."""

    def forward(self, logits, labels):
        return cross_entropy(logits, labels)


# ---------------------------------------------------------------- VAE losses

def vae_loss(recon_x, x, mu, logvar):
    """MSE(sum) reconstruction + analytic KL (federated_vae.py:97-108)."""
    mse = F.mse_loss(recon_x, x, reduction="sum")
    kld = -0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp())
    return mse + kld


def cost1(pk, px_z_mu, px_z_sig2, x):
    """E_qk{ -log p(x|theta) }: weighted Gaussian recon NLL
    (federated_vae_cl.py:101-109, batch loop vectorized)."""
    b = x.shape[0]
    err = (x - px_z_mu).pow(2) / (2 * px_z_sig2)
    err1 = 0.5 * torch.log(px_z_sig2 * 2 * math.pi)
    per_sample = (err + err1).reshape(b, -1).sum(dim=1)
    return (pk * per_sample).sum() / b


def cost2(pk):
    """Per-sample cluster entropy (federated_vae_cl.py:113-118)."""
    b = pk.shape[0]
    return -(pk * torch.log(pk + 1e-9)).sum() / b


def cost21(pk):
    """Inverse batch-entropy, prevents cluster collapse (federated_vae_cl.py:122-126)."""
    pbar = torch.mean(pk, 0)
    loss = -pbar * torch.log(pbar + 1e-9)
    return 1 / (loss + 1e-9)


def cost3(pk, q_z_mu, q_z_sig2, p_z_mu, p_z_sig2):
    """E_qk{ KL(q(z|x,k) || p(z|k)) } (federated_vae_cl.py:131-140)."""
    b = pk.shape[0]
    mudiff = (p_z_mu - q_z_mu).pow(2) / p_z_sig2
    sigratio = q_z_sig2 / p_z_sig2
    per_sample = (sigratio - torch.log(sigratio) + mudiff - 1).reshape(b, -1).sum(dim=1)
    return 0.5 * (pk * per_sample).sum() / b


def vaecl_loss(ekhat, mu_xi, sig2_xi, mu_b, sig2_b, mu_th, sig2_th, x,
               Kc=10, alpha=10.0, beta=1.0):
    """4-term clustering ELBO (federated_vae_cl.py:142-162)."""
    loss = 0
    for ci in range(Kc):
        pk = ekhat[:, ci]
        c1 = cost1(pk, mu_th[ci], sig2_th[ci], x)
        c2 = cost2(pk)
        c21 = cost21(pk)
        c3 = cost3(pk, mu_xi[ci], sig2_xi[ci], mu_b[ci], sig2_b[ci])
        loss = loss + c1 + alpha * (c2 + c3) + beta * c21
    return loss


# ------------------------------------------------------------------- InfoNCE

def info_nce(z: torch.Tensor, zhat: torch.Tensor) -> torch.Tensor:
    """InfoNCE over the patch grid (federated_cpc.py:149-180).

    z, zhat: [batch, channel, px, py].  Positive sample of patch (i) is the
    diagonal of the (px*py)^2 normalized inner-product matrix; negatives are
    the rest of its row.  One GEMM replaces the reference's O(p^4) loop.
    """
    assert z.shape == zhat.shape
    px, py = z.shape[2], z.shape[3]
    P = px * py
    Z = z.reshape(-1, P)
    Zhat = zhat.reshape(-1, P)
    Zn = Z / Z.norm(dim=0, keepdim=True)
    Zhatn = Zhat / Zhat.norm(dim=0, keepdim=True)
    zz = Zn.t() @ Zhatn                     # [P, P]
    softmax_diag = torch.softmax(zz, dim=1).diagonal()
    return -torch.log(softmax_diag + 1e-6).sum()
