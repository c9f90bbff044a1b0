#!/usr/bin/env python3
"""Per-block FedProx (proximal federated averaging) of K CIFAR10 clients
(reference parity: src/fedprox_multi.py; closure adds (rho/2)||x-z||^2,
z-update is the plain mean and is NOT written back into the clients).

torchrun --standalone --nproc-per-node K fedprox_multi.py --K K  for one
client per MI355X over RCCL/xGMI.
"""

from fedkit.parallel import FedConfig, FederatedJob
from fedkit.utils.cli import config_from_cli

# reference defaults (fedprox_multi.py:9-50)
K = 10
default_batch = 128
Nloop = 12
Nepoch = 1
Nadmm = 5
lambda1 = 0.0001
lambda2 = 0.0001
admm_rho0 = 1.0
load_model = False
init_model = True
save_model = True
check_results = True
biased_input = True
be_verbose = False
use_resnet = False
use_cuda = True


def main():
    cfg = config_from_cli(FedConfig(
        K=K, default_batch=default_batch, Nloop=Nloop, Nepoch=Nepoch,
        Nadmm=Nadmm, lambda1=lambda1, lambda2=lambda2, admm_rho0=admm_rho0,
        load_model=load_model, init_model=init_model, save_model=save_model,
        check_results=check_results, biased_input=biased_input,
        be_verbose=be_verbose, use_resnet=use_resnet, use_cuda=use_cuda,
        strategy="fedprox",
    ))
    FederatedJob(cfg).run()


if __name__ == "__main__":
    main()
