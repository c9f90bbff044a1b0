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

class _VaeElboFn(torch.autograd.Function):
    """MSE(sum) + analytic KLD as ONE fused reduction kernel
    (csrc/losses_extra.hip; SURVEY.md §2a 'single fused reduction kernel')."""

    @staticmethod
    def forward(ctx, recon, x, mu, logvar):
        from . import require_ext
        loss = require_ext().vae_elbo_fwd(recon, x, mu, logvar)
        ctx.save_for_backward(recon, x, mu, logvar)
        return loss

    @staticmethod
    def backward(ctx, gloss):
        from . import require_ext
        recon, x, mu, logvar = ctx.saved_tensors
        grecon, gmu, glogvar = require_ext().vae_elbo_bwd(
            recon, x, mu, logvar, gloss)
        return grecon, None, gmu, glogvar


def vae_loss(recon_x, x, mu, logvar):
    """MSE(sum) reconstruction + analytic KL (federated_vae.py:97-108)."""
    if _native(recon_x):
        if not (recon_x.dtype == x.dtype == mu.dtype == logvar.dtype):
            # mixed autocast dtypes: upcast so the fused kernel matches
            # the torch path's fp32 promotion exactly
            recon_x, x, mu, logvar = (t.float()
                                      for t in (recon_x, x, mu, logvar))
        return _VaeElboFn.apply(recon_x.contiguous(), x.contiguous(),
                                mu.contiguous(), logvar.contiguous())
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


class _VaeClTermsFn(torch.autograd.Function):
    """Per-(cluster, sample) cost1 / cost3 reductions in one kernel over
    [Kc*B] rows; backward is one elementwise pass (csrc/losses_extra.hip).
    R1[ci,b] is cost1's per-sample sum, R3[ci,b] is 2x cost3's."""

    @staticmethod
    def forward(ctx, x, mu_th, s_th, mu_q, s_q, mu_p, s_p, B):
        from . import require_ext
        R1, R3 = require_ext().vaecl_terms_fwd(x, mu_th, s_th, mu_q, s_q,
                                               mu_p, s_p, B)
        ctx.save_for_backward(x, mu_th, s_th, mu_q, s_q, mu_p, s_p)
        ctx.B = B
        return R1, R3

    @staticmethod
    def backward(ctx, gR1, gR3):
        from . import require_ext
        x, mu_th, s_th, mu_q, s_q, mu_p, s_p = ctx.saved_tensors
        outs = require_ext().vaecl_terms_bwd(
            x, mu_th, s_th, mu_q, s_q, mu_p, s_p,
            gR1.contiguous(), gR3.contiguous(), ctx.B)
        return (None, *outs, None)


def vaecl_loss(ekhat, mu_xi, sig2_xi, mu_b, sig2_b, mu_th, sig2_th, x,
               Kc=10, alpha=10.0, beta=1.0):
    """4-term clustering ELBO (federated_vae_cl.py:142-162).

    GPU path: the heavy cost1/cost3 per-sample reductions run as ONE HIP
    kernel over all Kc clusters (the reference loops over the batch in
    Python per cluster, 101-140); the tiny pk-weighted combines and
    cost2/cost21 stay in torch so ekhat's autograd is untouched.
    """
    if _native(x):
        B = x.shape[0]
        mu_th_s = torch.stack([mu_th[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        s_th_s = torch.stack([sig2_th[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        mu_q_s = torch.stack([mu_xi[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        s_q_s = torch.stack([sig2_xi[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        mu_p_s = torch.stack([mu_b[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        s_p_s = torch.stack([sig2_b[ci] for ci in range(Kc)]).reshape(Kc * B, -1)
        xf = x.reshape(B, -1)
        if not all(t.dtype == xf.dtype for t in
                   (mu_th_s, s_th_s, mu_q_s, s_q_s, mu_p_s, s_p_s)):
            xf = xf.float()
            mu_th_s, s_th_s, mu_q_s, s_q_s, mu_p_s, s_p_s = (
                t.float() for t in (mu_th_s, s_th_s, mu_q_s, s_q_s,
                                    mu_p_s, s_p_s))
        R1, R3 = _VaeClTermsFn.apply(xf.contiguous(), mu_th_s.contiguous(),
                                     s_th_s.contiguous(), mu_q_s.contiguous(),
                                     s_q_s.contiguous(), mu_p_s.contiguous(),
                                     s_p_s.contiguous(), B)
        pkT = ekhat.t().reshape(-1)                     # [Kc*B]
        c1_tot = (pkT * R1).sum() / B
        c3_tot = 0.5 * (pkT * R3).sum() / B
        c2_tot = -(ekhat * torch.log(ekhat + 1e-9)).sum() / B
        pbar = ekhat.mean(0)                            # [Kc]
        c21_tot = (1.0 / (-pbar * torch.log(pbar + 1e-9) + 1e-9)).sum()
        return c1_tot + alpha * (c2_tot + c3_tot) + beta * c21_tot
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

class _InfoNceFn(torch.autograd.Function):
    """Fused InfoNCE: column norms + similarity matrix + row softmax +
    -sum log(diag) in ONE kernel launch; backward one elementwise pass
    using t_i = sum_j dzz[i,j] zz[i,j] for the normalization chain
    (csrc/losses_extra.hip)."""

    @staticmethod
    def forward(ctx, Z, Zhat):
        from . import require_ext
        loss, zz, soft, norms = require_ext().info_nce_fwd(Z, Zhat)
        ctx.save_for_backward(Z, Zhat, zz, soft, norms)
        return loss

    @staticmethod
    def backward(ctx, gloss):
        from . import require_ext
        Z, Zhat, zz, soft, norms = ctx.saved_tensors
        P = zz.shape[0]
        # dloss/dzz[i,j] = -g * s_ii (delta_ij - s_ij) / (s_ii + 1e-6)
        diag = soft.diagonal()                          # [P]
        w = (diag / (diag + 1e-6)) * gloss              # [P]
        dzz = w.unsqueeze(1) * soft
        dzz.diagonal().sub_(w)                          # -(I - S) row-scaled
        t = (dzz * zz).sum(dim=1)                       # [P]
        u = (dzz * zz).sum(dim=0)                       # [P]
        tu = torch.stack([t, u])
        gZ, gZhat = require_ext().info_nce_bwd(Z, Zhat, dzz, tu, norms)
        return gZ, gZhat


def info_nce(z: torch.Tensor, zhat: torch.Tensor) -> torch.Tensor:
    """InfoNCE over the patch grid (federated_cpc.py:149-180).

    z, zhat: [batch, channel, px, py].  Positive sample of patch (i) is the
    diagonal of the (px*py)^2 normalized inner-product matrix; negatives are
    the rest of its row.  GPU: one fused kernel; CPU: one GEMM + row softmax
    (both replace the reference's O(p^4) Python loop, identical values).
    """
    assert z.shape == zhat.shape
    px, py = z.shape[2], z.shape[3]
    P = px * py
    Z = z.reshape(-1, P)
    Zhat = zhat.reshape(-1, P)
    if _native(z) and P <= 64:
        return _InfoNceFn.apply(Z.contiguous(), Zhat.contiguous())
    Zn = Z / Z.norm(dim=0, keepdim=True)
    Zhatn = Zhat / Zhat.norm(dim=0, keepdim=True)
    zz = Zn.t() @ Zhatn                     # [P, P]
    softmax_diag = torch.softmax(zz, dim=1).diagonal()
    return -torch.log(softmax_diag + 1e-6).sum()
