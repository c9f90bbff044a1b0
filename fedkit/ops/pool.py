"""NHWC pooling on the CDNA4 kernels (SURVEY.md §2a "MaxPool2d 2x2,
AvgPool2d 4x4 / 2x2" row).  GPU path: csrc/pool.hip — a pooling window in
channels_last is KH*KW coalesced vector loads of the same channel chunk;
max pool saves a per-output uint8 argmax so backward is one scatter pass.
CPU path: stock F.max_pool2d / F.avg_pool2d.
Non-overlapping (stride == kernel) only — the only form the reference's
models use (simple_models.py:13, 49-50, 89-92, 213, 464)."""

import torch
import torch.nn as nn
import torch.nn.functional as F


def _native(t):
    from . import native_enabled
    return native_enabled(t)


def _ext():
    from . import require_ext
    return require_ext()


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k):
        y, idx = _ext().max_pool2d_fwd(x, k)
        ctx.save_for_backward(idx)
        ctx.meta = (k, x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, gy):
        (idx,) = ctx.saved_tensors
        k, H, W = ctx.meta
        gy = gy.contiguous(memory_format=torch.channels_last)
        return _ext().max_pool2d_bwd(gy, idx, k, H, W), None


class _AvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k):
        ctx.meta = (k, x.shape[2], x.shape[3])
        return _ext().avg_pool2d_fwd(x, k)

    @staticmethod
    def backward(ctx, gy):
        k, H, W = ctx.meta
        gy = gy.contiguous(memory_format=torch.channels_last)
        return _ext().avg_pool2d_bwd(gy, k, H, W), None


def _pool_ok(x, k):
    return _native(x) and x.shape[2] % k == 0 and x.shape[3] % k == 0


def max_pool2d(x, k):
    if _pool_ok(x, k):
        return _MaxPoolFn.apply(
            x.contiguous(memory_format=torch.channels_last), k)
    return F.max_pool2d(x, k)


def avg_pool2d(x, k):
    if _pool_ok(x, k):
        return _AvgPoolFn.apply(
            x.contiguous(memory_format=torch.channels_last), k)
    return F.avg_pool2d(x, k)


class FedMaxPool2d(nn.MaxPool2d):
    """MaxPool2d(k, k) routed through the NHWC kernel on GPU."""

    def forward(self, x):
        k = self.kernel_size if isinstance(self.kernel_size, int) \
            else self.kernel_size[0]
        s = self.stride if isinstance(self.stride, int) else self.stride[0]
        if s == k and self.padding == 0 and self.dilation == 1:
            return max_pool2d(x, k)
        return super().forward(x)
