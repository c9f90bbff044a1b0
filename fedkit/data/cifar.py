"""CIFAR10 sharding + per-client biased normalization.

The reference shards 50,000 CIFAR10 train images into K contiguous ranges of
K_perslave = ceil(50000/K) and simulates non-IID clients with a per-client
biased Normalize((0.5+k/100, 0.5-k/100, 0.5), ...) (federated_multi.py:51-71).

This module reproduces that contract with two MI355X-first changes:
  * data lives ON DEVICE: each client's whole shard is resident in HBM
    (288 GB per GPU; a 1/K CIFAR shard is < 20 MB) and normalization is a
    device-side op, so the training loop never touches the host;
  * with no downloaded dataset available (this environment has no network),
    a deterministic synthetic CIFAR-shaped dataset stands in — same shapes,
    same sharding math, seeded identically on every rank.  Real CIFAR-10
    python batches (cifar-10-batches-py) are loaded when present.

Reference quirk: the range() construction drops the LAST index of every
shard (federated_multi.py:52-58).  Default here keeps all indices; pass
exact_reference_shards=True for bit-parity with the reference split.
"""

import os
import pickle
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

CIFAR_TRAIN_N = 50000
CIFAR_TEST_N = 10000


def shard_indices(K: int, n: int = CIFAR_TRAIN_N,
                  exact_reference_shards: bool = False) -> Dict[int, List[int]]:
    """Contiguous 1/K shards (federated_multi.py:52-58)."""
    per = (n + K - 1) // K
    shards = {}
    for ck in range(K):
        lo = per * ck
        hi = min(per * (ck + 1), n)
        if exact_reference_shards and per * (ck + 1) - 1 <= n:
            # reference's off-by-one: range(lo, per*(ck+1)-1)
            hi = min(per * (ck + 1) - 1, n)
        shards[ck] = list(range(lo, hi))
    return shards


def client_normalization(ck: int, biased_input: bool = True) -> Tuple[tuple, tuple]:
    """Per-client (mean, std) (federated_multi.py:60-71)."""
    if biased_input:
        return ((0.5 + ck / 100, 0.5 - ck / 100, 0.5),
                (0.5 + ck / 100, 0.5 - ck / 100, 0.5))
    return ((0.5, 0.5, 0.5), (0.5, 0.5, 0.5))


def _load_real_cifar(root: str):
    """Load cifar-10-batches-py if present (no torchvision dependency)."""
    base = os.path.join(root, "cifar-10-batches-py")
    if not os.path.isdir(base):
        return None
    def read(names):
        xs, ys = [], []
        for nm in names:
            with open(os.path.join(base, nm), "rb") as f:
                d = pickle.load(f, encoding="bytes")
            xs.append(np.asarray(d[b"data"], dtype=np.uint8))
            ys.extend(d[b"labels"])
        x = np.concatenate(xs).reshape(-1, 3, 32, 32)
        return torch.from_numpy(x), torch.tensor(ys, dtype=torch.long)
    try:
        train = read([f"data_batch_{i}" for i in range(1, 6)])
        test = read(["test_batch"])
        return train, test
    except (OSError, KeyError, pickle.UnpicklingError):
        return None


_SYNTH_CACHE = {}


def _synthetic_cifar(train: bool, seed: int = 1234):
    """Deterministic CIFAR-shaped uint8 images + labels.

    Class-STRUCTURED (not pure noise): one random smooth prototype per
    class plus heavy pixel noise, train/test drawn from the same
    distribution.  This keeps accuracy curves meaningful without network
    access to the real CIFAR10 — the federated-vs-standalone ORDERING the
    reference's comparison.png demonstrates (README.md:28-31) is
    reproducible on it (see profiles/acc_synthetic.md).
    """
    key = (train, seed)
    if key not in _SYNTH_CACHE:
        n = CIFAR_TRAIN_N if train else CIFAR_TEST_N
        M = 150                      # prototype clusters per class
        g = torch.Generator().manual_seed(seed)  # prototypes shared
        # smooth cluster prototypes: upsampled 8x8 random fields.  A sample
        # is one of its class's M clusters plus heavy noise, so accuracy is
        # limited by how many clusters the training shard has covered —
        # i.e. by data VOLUME, like a real vision task's learning curve.
        proto = torch.rand(10 * M, 3, 8, 8, generator=g)
        proto = torch.nn.functional.interpolate(proto, size=(32, 32),
                                                mode="bilinear",
                                                align_corners=False)
        proto = proto.view(10, M, 3, 32, 32)
        g2 = torch.Generator().manual_seed(seed + (0 if train else 1))
        y = torch.arange(n, dtype=torch.long) % 10
        perm = torch.randperm(n, generator=g2)
        y = y[perm]
        j = torch.randint(0, M, (n,), generator=g2)
        noise = torch.rand(n, 3, 32, 32, generator=g2)
        x = (0.4 * proto[y, j] + 0.6 * noise).mul_(255).to(torch.uint8)
        _SYNTH_CACHE[key] = (x, y)
    return _SYNTH_CACHE[key]


def load_cifar(root: str = "./torchdata", train: bool = True):
    """(images uint8 [N,3,32,32], labels int64 [N]); real data if present."""
    real = _load_real_cifar(root)
    if real is not None:
        return real[0] if train else real[1]
    return _synthetic_cifar(train)


class DeviceShardLoader:
    """Minibatch iterator over one client's HBM-resident shard.

    Equivalent to DataLoader(..., SubsetRandomSampler(shard)) in the
    reference (federated_multi.py:83): a fresh random permutation of the
    shard every epoch, batches of `batch` (last partial batch kept), images
    normalized with the client's biased transform ON DEVICE.
    """

    def __init__(self, images_u8: torch.Tensor, labels: torch.Tensor,
                 indices: Optional[List[int]], ck: int, batch: int,
                 device, biased_input: bool = True, shuffle: bool = True,
                 dtype: torch.dtype = torch.float32,
                 channels_last: bool = False,
                 generator: Optional[torch.Generator] = None):
        idx = torch.arange(images_u8.shape[0]) if indices is None \
            else torch.tensor(indices, dtype=torch.long)
        self.n = idx.numel()
        self.batch = batch
        self.device = device
        self.shuffle = shuffle
        self.generator = generator
        mean, std = client_normalization(ck, biased_input)
        m = torch.tensor(mean, device=device).view(1, 3, 1, 1)
        s = torch.tensor(std, device=device).view(1, 3, 1, 1)
        # whole shard resident on device, normalized once (x/255 - m)/s
        x = images_u8[idx].to(device=device, dtype=torch.float32) / 255.0
        x = (x - m) / s
        x = x.to(dtype)
        if channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        self.x = x
        self.y = labels[idx].to(device)

    def __len__(self):
        return (self.n + self.batch - 1) // self.batch

    def __iter__(self):
        if self.shuffle:
            order = torch.randperm(self.n, generator=self.generator)
        else:
            order = torch.arange(self.n)
        for i in range(0, self.n, self.batch):
            sel = order[i:i + self.batch].to(self.x.device)
            yield self.x[sel], self.y[sel]


def make_client_datasets(K: int, my_clients: List[int], batch: int, device,
                         biased_input: bool = True, root: str = "./torchdata",
                         exact_reference_shards: bool = False,
                         dtype: torch.dtype = torch.float32,
                         channels_last: bool = False,
                         shuffle_seed: int = 69):
    """(trainloader_dict, testloader_dict) for this rank's clients.

    Shuffling uses a per-client generator seeded from (shuffle_seed, ck) so
    the batch order of client ck is IDENTICAL whether it runs in-process
    (LocalComm) or as rank ck of a distributed job — the engine's
    local-vs-distributed bit-equality tests depend on this.
    """
    xtr, ytr = load_cifar(root, train=True)
    xte, yte = load_cifar(root, train=False)
    shards = shard_indices(K, xtr.shape[0], exact_reference_shards)
    train_loaders, test_loaders = {}, {}
    for ck in my_clients:
        g = torch.Generator()
        g.manual_seed(shuffle_seed * 100003 + ck)
        train_loaders[ck] = DeviceShardLoader(
            xtr, ytr, shards[ck], ck, batch, device, biased_input,
            shuffle=True, dtype=dtype, channels_last=channels_last,
            generator=g)
        test_loaders[ck] = DeviceShardLoader(
            xte, yte, None, ck, batch, device, biased_input,
            shuffle=False, dtype=dtype, channels_last=channels_last)
    return train_loaders, test_loaders
