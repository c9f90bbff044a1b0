"""FedLinear — nn.Linear whose GPU path is the fused HIP linear kernels.

Subclasses nn.Linear (state_dict / parameter-order parity with the
reference's fc layers, e.g. simple_models.py:15-17).  All fc layers in the
model zoo are tiny and latency-bound (SURVEY §2a "Linear" row, heads like
[128,512]x[512,10]); csrc/linear.hip runs each pass as ONE kernel with
fp32 accumulation and the bias fused into the forward, replacing the
rocBLAS GEMM + bias + (in backward) two more GEMMs dispatch chain.

CPU / non-bf16 path: stock F.linear (the numerics reference).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


def _native(t):
    from . import native_enabled
    return native_enabled(t)


def _ext():
    from . import require_ext
    return require_ext()


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias):
        y = _ext().linear_fwd(x, w, bias)
        ctx.save_for_backward(x, w)
        ctx.want_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        gy = gy.contiguous()
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = _ext().linear_bwd_data(gy, w)
        if ctx.needs_input_grad[1] or (ctx.want_bias and ctx.needs_input_grad[2]):
            outs = _ext().linear_bwd_weight(gy, x, ctx.want_bias)
            gw = outs[0].to(w.dtype)
            if ctx.want_bias:
                gb = outs[1]
        return gx, gw, gb


class FedLinear(nn.Linear):
    # Above this weight size a linear is a plain MFMA-shaped GEMM and the
    # library is the right tool (mandate: hand-written kernels for fused
    # hot ops, rocBLAS for plain GEMMs).  Net1 fc1 [512,1600] measured
    # 259 us on the wave-reduction bwd-weight kernel vs ~10 us library;
    # every head/VAE/Net-small fc sits far below the bound.
    _NATIVE_MAX_WEIGHT_NUMEL = 256 * 1024

    def forward(self, x):
        if _native(x) and self.weight.numel() <= self._NATIVE_MAX_WEIGHT_NUMEL:
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if x.dtype == torch.bfloat16 and x.dim() == 2:
                from .conv import _cached_frozen
                w16 = _cached_frozen(
                    self, "_wlin", self.weight,
                    lambda: self.weight.to(torch.bfloat16).contiguous())
                return _LinearFn.apply(x.contiguous(), w16, self.bias)
        return F.linear(x, self.weight.to(x.dtype),
                        self.bias.to(x.dtype) if self.bias is not None else None)
