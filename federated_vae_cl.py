#!/usr/bin/env python3
"""Federated variational-clustering VAE (reference parity:
src/federated_vae_cl.py; arXiv:2005.04613).

4-term clustering ELBO (cost1 + alpha*(cost2+cost3) + beta*cost21 summed
over Kc clusters), mixed optimizer schedule (Adam lr=1e-4 for the latent
block ci==2, stochastic LBFGS otherwise), reparametrization gated to the
latent block, always-on L2 regularization lambda2=1e-3.
"""

import functools

from fedkit.models import AutoEncoderCNNCL
from fedkit.optim import LBFGSNew
from fedkit.ops.losses import vaecl_loss
from fedkit.parallel import FedConfig, FederatedJob
from fedkit.utils.cli import config_from_cli

import torch.optim as optim

# reference defaults (federated_vae_cl.py:11-44)
K = 1
Kc = 10    # number of clusters
Lc = 32    # latent dimension
default_batch = 128
Nloop = 1
Nepoch = 1
Nadmm = 1
lambda2 = 0.001
load_model = False
init_model = True
save_model = True
biased_input = True
use_cuda = True


def main():
    cfg = config_from_cli(FedConfig(
        K=K, default_batch=default_batch, Nloop=Nloop, Nepoch=Nepoch,
        Nadmm=Nadmm, lambda1=0.0, lambda2=lambda2, l2_all_blocks=True,
        load_model=load_model, init_model=init_model, save_model=save_model,
        check_results=False, biased_input=biased_input, use_cuda=use_cuda,
        model="AutoEncoderCNNCL", strategy="fedavg",
    ))

    def loss_fn(net, images, _labels):
        ekhat, mu_xi, sig2_xi, mu_b, sig2_b, mu_th, sig2_th = net(images)
        return vaecl_loss(ekhat, mu_xi, sig2_xi, mu_b, sig2_b, mu_th,
                          sig2_th, images, Kc=Kc)

    def optimizer_factory(job, net, ci):
        # latent block gets Adam, everything else stochastic LBFGS
        # (federated_vae_cl.py:200-205)
        if ci == 2:
            return optim.Adam(
                filter(lambda p: p.requires_grad, net.parameters()), lr=1e-4)
        return LBFGSNew(filter(lambda p: p.requires_grad, net.parameters()),
                        history_size=10, max_iter=4, line_search_fn=True,
                        batch_mode=True)

    def block_hook(job, ci):
        # reparametrization only while training the latent block
        # (federated_vae_cl.py:185-189; note the reference's disable_repr is
        # a no-op quirk we preserve in the model class)
        for ck in job.comm.my_clients:
            if ci == 2:
                job.nets[ck].enable_repr()
            else:
                job.nets[ck].disable_repr()

    FederatedJob(cfg, model_factory=functools.partial(AutoEncoderCNNCL, K=Kc, L=Lc),
                 loss_fn=loss_fn, optimizer_factory=optimizer_factory,
                 block_hook=block_hook).run()


if __name__ == "__main__":
    main()
