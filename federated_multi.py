#!/usr/bin/env python3
"""Per-block Federated Averaging of K CIFAR10 clients (reference parity:
src/federated_multi.py).

Single process: all K clients simulated in-process (reference semantics).
Multi-GPU (one client per MI355X over RCCL/xGMI):
    torchrun --standalone --nproc-per-node K federated_multi.py --K K
Knobs keep the reference's names/defaults; override via CLI, e.g.
    python federated_multi.py --K 4 --Nloop 1 --use_resnet 1
"""

from fedkit.parallel import FedConfig, FederatedJob
from fedkit.utils.cli import config_from_cli

# reference defaults (federated_multi.py:9-48)
K = 10
default_batch = 128
Nloop = 12
Nepoch = 1
Nadmm = 3
lambda1 = 0.0001
lambda2 = 0.0001
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
        Nadmm=Nadmm, lambda1=lambda1, lambda2=lambda2,
        load_model=load_model, init_model=init_model, save_model=save_model,
        check_results=check_results, biased_input=biased_input,
        be_verbose=be_verbose, use_resnet=use_resnet, use_cuda=use_cuda,
        strategy="fedavg",
    ))
    FederatedJob(cfg).run()


if __name__ == "__main__":
    main()
