#!/usr/bin/env python3
"""Federated averaging of K convolutional VAEs on CIFAR10 (reference parity:
src/federated_vae.py; per-LAYER freezing via unfreeze_one_layer, Adam only,
VAE loss = MSE(sum) + analytic KLD, no test-set eval).
"""

from fedkit.ops.losses import vae_loss
from fedkit.parallel import FedConfig, FederatedJob
from fedkit.utils.cli import config_from_cli

# reference defaults (federated_vae.py:9-32)
K = 10
default_batch = 128
Nloop = 12
Nepoch = 1
Nadmm = 3
load_model = False
init_model = True
save_model = True
biased_input = True
use_cuda = True


def main():
    cfg = config_from_cli(FedConfig(
        K=K, default_batch=default_batch, Nloop=Nloop, Nepoch=Nepoch,
        Nadmm=Nadmm, load_model=load_model, init_model=init_model,
        save_model=save_model, check_results=False,
        biased_input=biased_input, use_cuda=use_cuda,
        model="AutoEncoderCNN", strategy="fedavg", per_layer=True,
        be_verbose=True,   # the reference prints every minibatch loss
    ))

    def loss_fn(net, images, _labels):
        out, mu, logvar = net(images)
        return vae_loss(out, images, mu, logvar)

    FederatedJob(cfg, loss_fn=loss_fn).run()


if __name__ == "__main__":
    main()
