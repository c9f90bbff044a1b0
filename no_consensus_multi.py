#!/usr/bin/env python3
"""K standalone CIFAR10 models, 1/K data each, NO communication — the
lower-bound baseline (reference parity: src/no_consensus_multi.py:
Adam lr=1e-3, 20 epochs, per-epoch eval, fresh optimizer per epoch).
"""

from fedkit.parallel import FedConfig
from fedkit.parallel.runtime import run_standalone
from fedkit.utils.cli import config_from_cli

# reference defaults (no_consensus_multi.py:9-37)
K = 10
default_batch = 128
Nepoch = 20
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
        K=K, default_batch=default_batch, Nloop=1, Nepoch=Nepoch, Nadmm=1,
        load_model=load_model, init_model=init_model, save_model=save_model,
        check_results=check_results, biased_input=biased_input,
        be_verbose=be_verbose, use_resnet=use_resnet, use_cuda=use_cuda,
        strategy="none",
    ))
    run_standalone(cfg)


if __name__ == "__main__":
    main()
