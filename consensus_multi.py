#!/usr/bin/env python3
"""Per-block consensus ADMM of K CIFAR10 clients with optional adaptive
(Barzilai-Borwein) penalty (reference parity: src/consensus_multi.py).

3-step ADMM: local minimization of f_k(x) + y_k^T(x-z) + (rho/2)||x-z||^2,
z = (1/(K rho)) sum(y_k + rho x_k) (ONE RCCL all-reduce over xGMI in the
distributed engine), y_k += rho (x_k - z).

torchrun --standalone --nproc-per-node K consensus_multi.py --K K  for one
client per MI355X.
"""

from fedkit.parallel import FedConfig, FederatedJob
from fedkit.utils.cli import config_from_cli

# reference defaults (consensus_multi.py:9-59)
K = 10
default_batch = 128
Nloop = 12
Nepoch = 1
Nadmm = 5
lambda1 = 0.0001
lambda2 = 0.0001
admm_rho0 = 0.1
load_model = False
init_model = True
save_model = True
check_results = True
biased_input = True
be_verbose = False
bb_update = False
bb_period_T = 2
bb_alphacorrmin = 0.2
bb_epsilon = 1e-3
bb_rhomax = 0.1
use_resnet = False
use_cuda = True


def main():
    cfg = config_from_cli(FedConfig(
        K=K, default_batch=default_batch, Nloop=Nloop, Nepoch=Nepoch,
        Nadmm=Nadmm, lambda1=lambda1, lambda2=lambda2, admm_rho0=admm_rho0,
        bb_update=bb_update, bb_period_T=bb_period_T,
        bb_alphacorrmin=bb_alphacorrmin, bb_epsilon=bb_epsilon,
        bb_rhomax=bb_rhomax,
        load_model=load_model, init_model=init_model, save_model=save_model,
        check_results=check_results, biased_input=biased_input,
        be_verbose=be_verbose, use_resnet=use_resnet, use_cuda=use_cuda,
        strategy="admm",
    ))
    FederatedJob(cfg).run()


if __name__ == "__main__":
    main()
