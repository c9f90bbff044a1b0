"""fedkit.ops — the CDNA4 HIP kernel library dispatch layer.

The native extension `fedkit._C` (built from csrc/ by __graft_entry__.build(),
in-tree, gfx950-only) provides the hand-written MI355X kernels:

  elu_fwd / elu_bwd              fused vectorized ELU (bf16/fp32)
  pack_params / unpack_params    multi-tensor flat pack/unpack (federation ABI)
  flat_axpy_params               p += alpha * flat_slice per parameter tensor
  conv2d_fwd / conv2d_bwd_*      NHWC implicit-GEMM conv on MFMA (3x3, 1x1)
  bn_fwd / bn_bwd                NHWC BatchNorm with fused ELU epilogue
  cross_entropy_fwd/bwd          fused log-softmax + NLL
  add_elu_fwd / add_elu_bwd      fused residual-add + ELU

Dispatch policy (this is the "native code is THE path that runs" rule):
  * on a ROCm GPU the HIP kernels are used; if the extension is missing the
    ops RAISE — there is no silent eager fallback on GPU;
  * on CPU (CI containers without a GPU) the ops fall back to stock torch so
    the full algorithm suite is testable without hardware.

Reference parity note: the reference has ZERO native code —
every kernel here replaces a cuDNN/cuBLAS dispatch listed in SURVEY.md §2a.
"""

import os

import torch

_EXT = None
_EXT_ERR = None


class _SyncDebugExt:
    """Kernel-fault localization (SURVEY §5 sanitizer gap, the MI355X
    analog of CUDA_LAUNCH_BLOCKING scoped to fedkit's own kernels):
    FEDKIT_SYNC_DEBUG=1 wraps every native call with a device synchronize
    + HIP error check, so an async fault (page fault, bad kernarg, OOB
    glds) surfaces AT the offending op with its name instead of crashing
    many launches later inside unrelated torch code."""

    def __init__(self, mod):
        self._mod = mod

    def __getattr__(self, name):
        fn = getattr(self._mod, name)
        if not callable(fn):
            return fn

        def wrapped(*a, **kw):
            out = fn(*a, **kw)
            if torch.cuda.is_available():
                try:
                    torch.cuda.synchronize()
                except RuntimeError as e:
                    raise RuntimeError(
                        f"fedkit._C.{name} faulted (FEDKIT_SYNC_DEBUG): {e}"
                    ) from e
            return out
        return wrapped


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        from .. import _C  # in-tree built extension: fedkit/_C*.so
        _EXT = _C
        if os.environ.get("FEDKIT_SYNC_DEBUG") == "1":
            _EXT = _SyncDebugExt(_C)
    except ImportError as e:  # pragma: no cover - exercised only pre-build
        _EXT_ERR = e


def ext():
    """Return the native extension module, or None when not built."""
    _try_load()
    return _EXT


def has_ext() -> bool:
    return ext() is not None


def require_ext():
    """Return the extension; raise loudly when running on GPU without it."""
    m = ext()
    if m is None:
        raise RuntimeError(
            "fedkit HIP extension (fedkit._C) is not built but a GPU op was "
            "requested on a ROCm device. Build it with "
            "`python __graft_entry__.py build` (hipcc --offload-arch=gfx950). "
            f"Import error: {_EXT_ERR!r}"
        )
    return m


# Kill switch for A/B benchmarking only (FEDKIT_NATIVE=0 forces eager torch).
_NATIVE_ENV = os.environ.get("FEDKIT_NATIVE", "1") != "0"


def native_enabled(t: torch.Tensor) -> bool:
    """True when tensor t should take the hand-written HIP path."""
    return _NATIVE_ENV and t.is_cuda


from . import elu, flat, losses          # noqa: E402,F401
from .conv import (FedConv2d, FedConvGeneric,  # noqa: E402
                   FedConvTranspose2d)
from .pool import FedMaxPool2d, max_pool2d, avg_pool2d  # noqa: E402
from .norm import FedBatchNorm2d         # noqa: E402
from .linear import FedLinear            # noqa: E402

__all__ = [
    "ext", "has_ext", "require_ext", "native_enabled",
    "elu", "flat", "losses", "FedConv2d", "FedConvGeneric",
    "FedMaxPool2d", "max_pool2d", "avg_pool2d",
    "FedConvTranspose2d", "FedBatchNorm2d", "FedLinear",
]
