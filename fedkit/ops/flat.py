"""Flat parameter-vector ops — the federation / optimizer data plane.

The reference flattens trainable parameters with per-tensor copies inside
Python loops (simple_utils.py:47-77; lbfgsnew.py:81-121).  On MI355X each of
those is its own tiny kernel launch; here the whole pack/unpack/axpy runs as
ONE HIP kernel over a device-side descriptor table (csrc/flat_ops.hip), and
the CPU fallback uses torch._foreach_* multi-tensor ops.

API (all fp32, used by fedkit.utils.paramvec, fedkit.optim.LBFGSNew and the
aggregation strategies):
  pack(tensors)            -> new flat fp32 vector [sum numel]
  pack_into(tensors, flat) -> fills an existing flat vector
  unpack(flat, tensors)    -> scatters flat back into the tensors
  add_flat(tensors, flat, alpha) -> t += alpha * flat_slice  per tensor
"""

from typing import List, Sequence

import torch


def _use_native(ts: Sequence[torch.Tensor]) -> bool:
    from . import native_enabled
    return len(ts) > 0 and native_enabled(ts[0])


def total_numel(tensors: Sequence[torch.Tensor]) -> int:
    return sum(t.numel() for t in tensors)


def _flat_views(flat: torch.Tensor, tensors: Sequence[torch.Tensor]) -> List[torch.Tensor]:
    """Views of `flat` shaped like each tensor, laid out in each tensor's
    PHYSICAL (storage) element order — matching what the native
    pack/unpack/axpy kernels traverse (they walk storage linearly).  For a
    channels_last 4D tensor the slice is viewed [N,H,W,C] then permuted to
    the logical [N,C,H,W] shape, so elementwise copies between the view and
    the tensor land in storage order.  Keeps the CPU fallback bit-consistent
    with GPU-produced federation vectors."""
    views = []
    off = 0
    for t in tensors:
        n = t.numel()
        sl = flat.narrow(0, off, n)
        if t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last) \
                and not t.is_contiguous():
            N, C, H, W = t.shape
            views.append(sl.view(N, H, W, C).permute(0, 3, 1, 2))
        else:
            views.append(sl.view_as(t))
        off += n
    if off != flat.numel():
        raise ValueError(f"flat vector has {flat.numel()} elements, tensors need {off}")
    return views


def _bump_versions(tensors: Sequence[torch.Tensor]) -> None:
    """The native kernels write through raw data pointers, bypassing the
    dispatcher's in-place version bump; derived-weight caches key on
    `param._version` (fedkit.ops.conv._cached_frozen), so a silent write
    would serve stale frozen bf16 weights after put_trainable_values /
    LBFGS moves.  `.data`/`.detach()` share the counter
    with the parameter, so bumping here covers the param itself."""
    for t in tensors:
        torch.autograd.graph.increment_version(t)


def pack_into(tensors: Sequence[torch.Tensor], flat: torch.Tensor) -> torch.Tensor:
    if _use_native(tensors):
        from . import require_ext
        require_ext().pack_params(list(tensors), flat)
        return flat
    torch._foreach_copy_(_flat_views(flat, tensors), list(tensors))
    return flat


def pack(tensors: Sequence[torch.Tensor], device=None) -> torch.Tensor:
    n = total_numel(tensors)
    dev = device if device is not None else (tensors[0].device if tensors else "cpu")
    flat = torch.empty(n, dtype=torch.float32, device=dev)
    return pack_into(tensors, flat)


def unpack(flat: torch.Tensor, tensors: Sequence[torch.Tensor]) -> None:
    if _use_native(tensors):
        from . import require_ext
        require_ext().unpack_params(flat, list(tensors))
        _bump_versions(tensors)
        return
    torch._foreach_copy_(list(tensors), _flat_views(flat, tensors))


def add_flat(tensors: Sequence[torch.Tensor], flat: torch.Tensor, alpha: float) -> None:
    """t += alpha * flat_slice for each tensor (the LBFGS _add_grad step)."""
    if _use_native(tensors):
        from . import require_ext
        require_ext().add_flat_params(list(tensors), flat, float(alpha))
        _bump_versions(tensors)
        return
    torch._foreach_add_(list(tensors), _flat_views(flat, tensors), alpha=alpha)


def fused_available(t: torch.Tensor) -> bool:
    """True when the multi_dot/lincomb HIP kernels can run on t."""
    return _use_native([t])


def multi_dot(vecs: Sequence[torch.Tensor], x: torch.Tensor) -> torch.Tensor:
    """x . vecs[i] for all i in ONE pass over x (device fp32 [len(vecs)]).

    Replaces the reference two-loop's serial torch dots (lbfgsnew.py:645-659),
    each of which is a separate launch plus a device->host sync.
    """
    if _use_native([x]):
        from . import require_ext
        return require_ext().multi_dot(list(vecs), x)
    return torch.stack([torch.dot(v, x) for v in vecs])


def lincomb(g: torch.Tensor, cg: float, vecs: Sequence[torch.Tensor],
            coeffs: Sequence[float]) -> torch.Tensor:
    """cg*g + sum coeffs[i]*vecs[i] in one fused pass (direction build)."""
    if _use_native([g]):
        from . import require_ext
        return require_ext().lincomb(g, float(cg), list(vecs),
                                     [float(c) for c in coeffs])
    out = g.mul(cg)
    for c, v in zip(coeffs, vecs):
        out.add_(v, alpha=float(c))
    return out
