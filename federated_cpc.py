#!/usr/bin/env python3
"""Federated Contrastive Predictive Coding on LOFAR visibilities
(reference parity: src/federated_cpc.py).

K clients, each one (HDF5 file, sub-array pointing) pair; encoder /
context-generator / predictor trained round-robin (the other two frozen),
stochastic LBFGS (history 7, max_iter 2), InfoNCE loss over the 32x32 patch
grid, per-sub-model FedAvg.

Fixes vs the reference (documented):
  * unfreeze_one_block(net, Bi[ci]) passed the [lo,hi] PAIR where an index
    is expected -> TypeError in the reference; fedkit's unfreeze_one_block
    accepts both forms;
  * InfoNCE is one GEMM + row-softmax instead of the O((px*py)^2) Python
    double loop of torch.dot (identical values);
  * load_model defaults False (the reference requires pre-existing
    checkpoints by default);
  * without h5py / the LOFAR files, a seeded synthetic visibility generator
    with the real reader's shapes stands in (fedkit.data.lofar).
"""

import argparse

import numpy as np
import torch

from fedkit.data import lofar_minibatch
from fedkit.models import ContextgenCNN, EncoderCNN, PredictorCNN
from fedkit.optim import LBFGSNew
from fedkit.ops.losses import info_nce
from fedkit.parallel.comm import make_comm
from fedkit.utils import (freeze_all_layers, get_trainable_values,
                          init_weights, put_trainable_values,
                          unfreeze_one_block)
from fedkit.utils.checkpoint import save_submodel, load_submodel

# reference defaults (federated_cpc.py:18-40, 137-145)
K = 4
Lc = 256     # latent dimension
Rc = 32      # reduced latent dimension
batch_size = 128
Nloop = 1
Niter = 10
Nadmm = 1
load_model = False
init_model = True
save_model = True
be_verbose = True
use_cuda = True

SUBMODELS = ("encoder", "contextgen", "predictor")


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--K", type=int, default=K)
    ap.add_argument("--Lc", type=int, default=Lc)
    ap.add_argument("--Rc", type=int, default=Rc)
    ap.add_argument("--batch_size", type=int, default=batch_size)
    ap.add_argument("--Nloop", type=int, default=Nloop)
    ap.add_argument("--Niter", type=int, default=Niter)
    ap.add_argument("--Nadmm", type=int, default=Nadmm)
    ap.add_argument("--load_model", type=int, default=int(load_model))
    ap.add_argument("--init_model", type=int, default=int(init_model))
    ap.add_argument("--save_model", type=int, default=int(save_model))
    ap.add_argument("--use_cuda", type=int, default=int(use_cuda))
    ap.add_argument("--file_list", type=str, nargs="*", default=None,
                    help="K LOFAR .h5 files (synthetic data when omitted)")
    ap.add_argument("--sap_list", type=str, nargs="*", default=None)
    args, _ = ap.parse_known_args(argv)

    comm = make_comm(args.K, device=None if args.use_cuda else torch.device("cpu"))
    device = comm.device() if args.use_cuda else torch.device("cpu")
    files = args.file_list or [None] * args.K
    saps = args.sap_list or ["0"] * args.K
    assert len(files) == args.K and len(saps) == args.K

    nets = {}
    for ck in comm.my_clients:
        nets[ck] = {
            "encoder": EncoderCNN(latent_dim=args.Lc).to(device),
            "contextgen": ContextgenCNN(latent_dim=args.Lc).to(device),
            "predictor": PredictorCNN(latent_dim=args.Lc, reduced_dim=args.Rc).to(device),
        }
        if args.load_model:
            for name in SUBMODELS:
                load_submodel(nets[ck][name], name, device)
    if args.init_model and not args.load_model:
        for ck in comm.my_clients:
            torch.manual_seed(0)
            for name in SUBMODELS:
                nets[ck][name].apply(init_weights)

    rngs = {ck: np.random.RandomState(1000 + ck) for ck in comm.my_clients}

    for nloop in range(args.Nloop):
        for mdl, name in enumerate(SUBMODELS):
            Bi = nets[comm.my_clients[0]][name].train_order_block_ids()
            for ck in comm.my_clients:
                for other in SUBMODELS:
                    if other != name:
                        freeze_all_layers(nets[ck][other])
            for ci in range(len(Bi)):
                for ck in comm.my_clients:
                    unfreeze_one_block(nets[ck][name], Bi[ci])
                N = sum(p.numel() for p in nets[comm.my_clients[0]][name].parameters()
                        if p.requires_grad)
                z = torch.zeros(N, dtype=torch.float32, device=device)
                opts = {ck: LBFGSNew(
                    filter(lambda p: p.requires_grad, nets[ck][name].parameters()),
                    history_size=7, max_iter=2, line_search_fn=True,
                    batch_mode=True) for ck in comm.my_clients}

                for nadmm in range(args.Nadmm):
                    for ck in comm.my_clients:
                        trio = nets[ck]
                        for niter in range(args.Niter):
                            patchx, patchy, y = lofar_minibatch(
                                filename=files[ck], batch_size=args.batch_size,
                                SAP=saps[ck], device=device, rng=rngs[ck],
                                synthetic_seed=ck)

                            def closure():
                                if torch.is_grad_enabled():
                                    opts[ck].zero_grad()
                                out = trio["encoder"](y)
                                # [B*px*py, latent] -> [B, px, py, latent]
                                # (same view order as federated_cpc.py:260-264)
                                out = out.contiguous().view(
                                    args.batch_size, patchx, patchy, -1)
                                latents = out.permute([0, 3, 1, 2]).contiguous()
                                context = trio["contextgen"](latents)
                                reduced_latents, prediction = trio["predictor"](latents, context)
                                loss = info_nce(reduced_latents, prediction)
                                if loss.requires_grad:
                                    loss.backward()
                                    if be_verbose and comm.is_primary:
                                        print('%d %d %d %f' % (nadmm, ck, niter,
                                                               loss.data.item()))
                                return loss

                            opts[ck].step(closure)
                            del y

                    # per-sub-model FedAvg (federated_cpc.py:280-304)
                    x = {ck: get_trainable_values(nets[ck][name], device)
                         for ck in comm.my_clients}
                    znew = comm.sum_across_clients({k: v.clone() for k, v in x.items()})
                    znew /= comm.K
                    dual_residual = torch.norm(z - znew).item() / N
                    if comm.is_primary:
                        print('dual (N=%d,iter=%d,loop=%d,model=%d,block=%d,avg=%d)=%e'
                              % (N, args.Niter - 1, nloop, mdl, ci, nadmm, dual_residual))
                    z = znew
                    for ck in comm.my_clients:
                        put_trainable_values(nets[ck][name], z)

    if args.save_model:
        for ck in comm.my_clients:
            for name in SUBMODELS:
                save_submodel(nets[ck][name], name, k=ck)


if __name__ == "__main__":
    main()
