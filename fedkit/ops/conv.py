"""FedConv2d — conv2d whose GPU path is a hand-written CDNA4 implicit-GEMM.

Subclasses nn.Conv2d so parameter order, state_dict layout and init match the
reference's modules exactly (the federated block partitions index parameter
tensors by position — simple_models.py:222-226).

GPU path (ROCm, fedkit._C built): NHWC bf16 implicit-GEMM on MFMA
(csrc/conv2d_mfma.hip) for the ResNet shapes (3x3 s1/s2 pad1 and 1x1 s1/s2,
no bias), covering fwd, bwd-data and bwd-weight.
fp32 accumulate; weights/activations bf16.  Unsupported shapes raise on GPU
rather than silently falling back (bias convs in the non-flagship models use
plain nn.Conv2d modules and are not routed here).

CPU path: stock F.conv2d (tests compare the HIP kernels against this).
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


class _ConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride, padding):
        # x: [N,C,H,W] logical, carried NHWC-contiguous; w: [K,C,R,S].
        # The padded input is computed ONCE and saved, so bwd-weight reuses
        # it instead of re-padding (and x itself need not be kept).
        ext = _ext()
        small_c = x.shape[1] % 8 != 0
        if small_c:
            y = ext.conv2d_fwd(x, w, stride, padding)
            ctx.save_for_backward(x, w)
        else:
            xp = ext.conv2d_pad_input(x, padding)
            y = ext.conv2d_fwd_prepadded(xp, w, stride)
            ctx.save_for_backward(xp, w)
        ctx.small_c = small_c
        ctx.stride = stride
        ctx.padding = padding
        ctx.hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, gy):
        xsaved, w = ctx.saved_tensors
        gy = gy.contiguous(memory_format=torch.channels_last)
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = _ext().conv2d_bwd_data(gy, w, ctx.stride, ctx.padding,
                                        ctx.hw[0], ctx.hw[1])
        if ctx.needs_input_grad[1]:
            if ctx.small_c:
                gw = _ext().conv2d_bwd_weight(gy, xsaved, ctx.stride,
                                              ctx.padding, w.shape[2], w.shape[3])
            else:
                gw = _ext().conv2d_bwd_weight_prepadded(
                    gy, xsaved, ctx.stride, w.shape[2], w.shape[3])
        return gx, gw, None, None


class FedConv2d(nn.Conv2d):
    def forward(self, x):
        if _native(x) and self.bias is None and self.groups == 1 \
                and self.dilation == (1, 1) and self.kernel_size[0] in (1, 3):
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if x.dtype == torch.bfloat16:
                # the MI355X fast path: bf16 NHWC through the MFMA kernels;
                # the fp32->bf16 weight cast is autograd-tracked, so bwd-weight
                # gradients land on the fp32 master copy
                x = x.contiguous(memory_format=torch.channels_last)
                w = self.weight.to(torch.bfloat16).contiguous(
                    memory_format=torch.channels_last)
                return _ConvFn.apply(x, w, self.stride[0], self.padding[0])
        return F.conv2d(x, self.weight.to(x.dtype),
                        self.bias.to(x.dtype) if self.bias is not None else None,
                        self.stride, self.padding, self.dilation, self.groups)
