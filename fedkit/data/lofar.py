"""LOFAR visibility reader + device-side patching for the CPC path.

Reference: federated_cpc.py:52-108 (get_data_minibatch) reads HDF5
measurement/saps/{SAP}/visibilities [nbase, ntime, nfreq, npol, 2] with
per-baseline scale factors, builds a [batch, 8, ntime, nfreq] tensor
(4 pol x {re, im}), unfolds into 32x32 patches with stride 16, and clamps
to +-1e6.

MI355X changes:
  * the unfold/patch reordering happens ON DEVICE in one reshape/permute
    (the reference copies patch-by-patch in a Python loop);
  * the per-baseline scaling is a broadcast multiply on device;
  * h5py is optional (not installed in the CI container) — a seeded
    synthetic visibility generator with the same shapes stands in so the
    CPC driver and tests run anywhere.
"""

from typing import Optional, Tuple

import numpy as np
import torch

try:
    import h5py  # noqa: F401
    HAS_H5PY = True
except ImportError:
    HAS_H5PY = False


def synthetic_visibilities(nbase=64, ntime=128, nfreq=128, npol=4, seed=0):
    """CIFAR-of-LOFAR: deterministic complex visibilities + scale factors."""
    g = torch.Generator().manual_seed(seed)
    vis = torch.randn(nbase, ntime, nfreq, npol, 2, generator=g)
    # heavy-tailed amplitudes like real visibilities
    vis = vis * torch.exp(2.0 * torch.randn(nbase, 1, 1, npol, 1, generator=g))
    scale = torch.rand(nbase, nfreq, npol, generator=g) + 0.5
    return vis, scale


def patch_visibilities(x: torch.Tensor, patch_size: int = 32) -> Tuple[int, int, torch.Tensor]:
    """[B, 8, ntime, nfreq] -> (patchx, patchy, [B*px*py, 8, p, p]).

    Stride = patch_size//2 (half overlap), patch-major ordering identical to
    the reference's copy loop (federated_cpc.py:84-99): output row block k
    holds patch (k // py, k % py) of every baseline.
    """
    stride = patch_size // 2
    y = x.unfold(2, patch_size, stride).unfold(3, patch_size, stride)
    # y: [B, C, px, py, p, p] -> [px*py, B, C, p, p] -> flat
    b, c, px, py, p1, p2 = y.shape
    y = y.permute(2, 3, 0, 1, 4, 5).reshape(px * py * b, c, p1, p2)
    return px, py, y.contiguous()


def _assemble(vis: torch.Tensor, scale: torch.Tensor, baselines,
              device) -> torch.Tensor:
    """[B,8,ntime,nfreq] with per-baseline frequency scaling applied."""
    v = vis[baselines].to(device)          # [B, ntime, nfreq, 4, 2]
    s = scale[baselines].to(device)        # [B, nfreq, 4]
    v = v * s[:, None, :, :, None]
    # interleave (re, im) per polarization -> 8 channels
    b, nt, nf, npol, _ = v.shape
    v = v.permute(0, 3, 4, 1, 2).reshape(b, npol * 2, nt, nf)
    return v


def lofar_minibatch(filename: Optional[str] = None, batch_size: int = 2,
                    patch_size: int = 32, SAP: str = "0", device="cpu",
                    rng: Optional[np.random.RandomState] = None,
                    synthetic_seed: int = 0):
    """(patchx, patchy, y) with y = [batch*px*py, 8, p, p], clamped +-1e6.

    With a filename and h5py available, reads the real LOFAR H5 layout;
    otherwise generates synthetic visibilities of the same shape.
    """
    rng = rng or np.random
    if filename is not None and HAS_H5PY:
        import h5py
        with h5py.File(filename, "r") as f:
            g = f["measurement"]["saps"][SAP]["visibilities"]
            h = f["measurement"]["saps"][SAP]["visibility_scale_factors"]
            nbase = g.shape[0]
            baselines = np.sort(rng.randint(0, nbase, batch_size))
            vis = torch.from_numpy(np.asarray(g[baselines], dtype=np.float32))
            scale = torch.from_numpy(np.asarray(h[baselines], dtype=np.float32))
        x = _assemble(vis, scale, slice(None), device)
    else:
        vis, scale = synthetic_visibilities(seed=synthetic_seed)
        baselines = rng.randint(0, vis.shape[0], batch_size)
        x = _assemble(vis, scale, baselines, device)
    px, py, y = patch_visibilities(x, patch_size)
    y.clamp_(-1e6, 1e6)
    return px, py, y
