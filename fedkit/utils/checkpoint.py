"""Checkpoint save/load with the reference's on-disk contract.

Per-client file ./s{k}.model is a torch.save dict with keys
{model_state_dict, epoch, optimizer_state_dict, running_loss}
(reference federated_multi.py:226-233); CPC saves per-sub-model files
encoder{k}.model / contextgen{k}.model / predictor{k}.model with
model_state_dict only (federated_cpc.py:308-318).  This layout is the
compatibility contract — a checkpoint written by the
reference loads here and vice versa.
"""

import os

import torch


def client_ckpt_path(k: int, prefix: str = "./s") -> str:
    return f"{prefix}{k}.model"


def save_client_checkpoint(net, opt, epoch, running_loss, k, prefix="./s"):
    torch.save({
        "model_state_dict": net.state_dict(),
        "epoch": epoch,
        "optimizer_state_dict": opt.state_dict() if opt is not None else {},
        "running_loss": running_loss,
    }, client_ckpt_path(k, prefix))


def load_client_checkpoint(net, k, mydevice=None, prefix="./s", train=True):
    path = client_ckpt_path(k, prefix)
    ckpt = torch.load(path, map_location=mydevice, weights_only=False)
    net.load_state_dict(ckpt["model_state_dict"])
    if train:
        net.train()
    return ckpt


def save_submodel(net, name: str, k=None):
    """CPC layout: '<name>{k}.model' with model_state_dict only."""
    suffix = "" if k is None else str(k)
    torch.save({"model_state_dict": net.state_dict()}, f"{name}{suffix}.model")


def load_submodel(net, name: str, mydevice=None, k=None, train=True):
    suffix = "" if k is None else str(k)
    path = f"{name}{suffix}.model"
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    ckpt = torch.load(path, map_location=mydevice, weights_only=False)
    net.load_state_dict(ckpt["model_state_dict"])
    if train:
        net.train()
    return ckpt
