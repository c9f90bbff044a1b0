"""ELU activation: hand-written vectorized HIP kernel on GPU, F.elu on CPU.

The reference applies F.elu after every conv/fc (e.g. simple_models.py:20-24,
150-153); on MI355X this is a pure HBM-bandwidth op, so the kernel loads
bf16 as 8-wide vectors (guide G13) and the backward uses only the saved
OUTPUT (dx = dy * (y > 0 ? 1 : y + 1) for alpha=1), halving saved memory.
"""

import torch
import torch.nn.functional as F


def _ext():
    from . import require_ext
    return require_ext()


def _native(t):
    from . import native_enabled
    return native_enabled(t)


class _EluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y = _ext().elu_fwd(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, gy):
        (y,) = ctx.saved_tensors
        return _ext().elu_bwd(gy, y)


def elu(x: torch.Tensor) -> torch.Tensor:
    if _native(x):
        return _EluFn.apply(x)
    return F.elu(x)
