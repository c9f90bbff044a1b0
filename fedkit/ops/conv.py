"""FedConv2d — conv2d whose GPU path is a hand-written CDNA4 implicit-GEMM.

Subclasses nn.Conv2d so parameter order, state_dict layout and init match the
reference's modules exactly (the federated block partitions index parameter
tensors by position — simple_models.py:222-226).

GPU path (ROCm, fedkit._C built): NHWC bf16 implicit-GEMM on MFMA
(csrc/conv2d_mfma.hip), fp32 accumulate, covering fwd, bwd-data and
bwd-weight:
  * FedConv2d — the flagship ResNet shapes (3x3 s1/s2 pad1, 1x1 s1/s2,
    no bias); unsupported shapes raise on GPU rather than silently
    falling back;
  * FedConvGeneric — any channel counts via zero channel padding
    (in -> x8, out -> x64 with dense-at-true-K stores), bias, dilation:
    the VAE encoder 4x4-s2 chain and the CPC dilated bank / 2x2 / 1x1
    convs;
  * FedConvTranspose2d — the VAE decoder deconvs as dilate-pad + the
    stride-1 conv adjoint.

CPU path: stock F.conv2d (tests compare the HIP kernels against this).
"""

import os

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
    def forward(ctx, x, w, stride, padding, pbuf=None):
        # x: [N,C,H,W] logical, carried NHWC-contiguous; w: [K,C,R,S].
        # The padded input is computed ONCE and saved, so bwd-weight reuses
        # it instead of re-padding (and x itself need not be kept).
        # pbuf (round 2): the producing bn_elu(pad_out=...) already wrote
        # the PADDED image; x is its interior view (the autograd edge).
        # The forward and bwd-weight consume pbuf directly (no pad pass);
        # bwd-data computes the input gradient at the ORIGINAL unpadded
        # geometry, so no border-gradient work is wasted.
        # FEDKIT_CONV_BNSTATS=1: the epilogue also emits the BatchNorm
        # stage-1 partials and the downstream FedBatchNorm2d skips its own
        # reduction pass.  MEASURED NET LOSS at CIFAR sizes (step 3.9 ->
        # ~4.9 ms): the per-channel finalize must then read one partial
        # row per conv workgroup with XCD-crossing strided loads, which
        # costs more than the single streaming pass it replaces — kept as
        # a documented experiment for larger-image workloads.
        ext = _ext()
        small_c = x.shape[1] % 8 != 0
        want_stats = os.environ.get("FEDKIT_CONV_BNSTATS") == "1"
        if small_c:
            y = ext.conv2d_fwd(x, w, stride, padding)
            part = y.new_empty(0)
            ctx.save_for_backward(x, w)
        elif pbuf is not None:
            y = ext.conv2d_fwd_prepadded(pbuf, w, stride)
            part = y.new_empty(0)
            ctx.save_for_backward(pbuf, w)
        else:
            xp = ext.conv2d_pad_input(x, padding)
            if want_stats:
                y, part = ext.conv2d_fwd_prepadded_bnstats(xp, w, stride)
            else:
                y = ext.conv2d_fwd_prepadded(xp, w, stride)
                part = y.new_empty(0)
            ctx.save_for_backward(xp, w)
        ctx.small_c = small_c
        ctx.stride = stride
        ctx.padding = padding
        ctx.hw = (x.shape[2], x.shape[3])
        ctx.mark_non_differentiable(part)
        return y, part

    @staticmethod
    def backward(ctx, gy, _gpart):
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
        return gx, gw, None, None, None


def _pad_mult(n, m):
    return (n + m - 1) // m * m


def _cached_frozen(mod, key, src_param, make):
    """Per-module cache of a derived (bf16/padded) weight while the source
    parameter is FROZEN.  In per-block federated training all but one
    layer-block is frozen (simple_utils.py:34-45 semantics), so the
    per-forward fp32->bf16 cast of every conv weight (~200 us/step on
    ResNet18) is recomputed only when the parameter's in-place version
    counter moves (optimizer step / put_trainable_values / load).
    Trainable parameters are never cached: their cast must live in the
    current autograd graph so gradients reach the fp32 master."""
    if src_param.requires_grad:
        return make()
    ver = src_param._version
    cache = getattr(mod, key, None)
    if cache is not None and cache[0] == ver:
        return cache[1]
    with torch.no_grad():
        out = make()
    setattr(mod, key, (ver, out))
    return out


class _BatchCastFn(torch.autograd.Function):
    """Fused fp32->bf16 cast of MANY weight tensors in one kernel, grads
    cast back in one kernel (csrc/flat_ops.hip cast tables).  A full-model
    training step otherwise pays one cast launch per conv each way
    (~40 launches, ~200 us/step on ResNet18)."""

    @staticmethod
    def forward(ctx, *ws):
        outs = tuple(torch.empty_like(w, dtype=torch.bfloat16) for w in ws)
        _ext().cast_f32_to_bf16(list(ws), list(outs))
        return outs

    @staticmethod
    def backward(ctx, *gys):
        outs, srcs, dsts = [], [], []
        for g in gys:
            if g is None:
                outs.append(None)
            else:
                f = torch.empty_like(g, dtype=torch.float32)
                srcs.append(g)
                dsts.append(f)
                outs.append(f)
        if srcs:
            _ext().cast_bf16_to_f32(srcs, dsts)
        return tuple(outs)


def batch_cast_weights(mods):
    """One-kernel bf16 cast of the given modules' weights; each module's
    next native forward consumes its copy (version-checked one-shot)."""
    if not mods:
        return
    outs = _BatchCastFn.apply(*[m.weight for m in mods])
    for m, o in zip(mods, outs):
        m._w16_once = (m.weight._version, o)


class _GenConvFn(torch.autograd.Function):
    """Generic conv on the MFMA kernels: any in/out channel count via zero
    channel padding (in -> mult of 8, out -> mult of 64, with the kernel
    storing dense at the true K), square filter, stride 1/2, dilation.

    Covers the VAE encoder/decoder 4x4-s2 convs, the CPC dilated conv bank
    (d in 1..16) and the CPC 2x2/1x1 convs (SURVEY.md §2a rows 3, 6;
    BASELINE north star: "the VAE/CPC encoder convs — hand-written CDNA4").
    Inputs arrive already channel-padded (xpch [N][C8][H][W] bf16 NHWC,
    wp [K64][C8][R][S]); the callers pad via F.pad so autograd slices the
    gradients back down.
    """

    @staticmethod
    def forward(ctx, xpch, wp, stride, padding, dil, ktrue):
        ext = _ext()
        xp = ext.conv2d_pad_input(xpch, padding)
        y = ext.conv2d_fwd_prepadded(xp, wp, stride, dil=dil, ktrue=ktrue)
        ctx.save_for_backward(xp, wp)
        ctx.meta = (stride, padding, dil, ktrue,
                    xpch.shape[2], xpch.shape[3])
        return y

    @staticmethod
    def backward(ctx, gy):
        xp, wp = ctx.saved_tensors
        stride, padding, dil, ktrue, H, W = ctx.meta
        K64 = wp.shape[0]
        gy = gy.contiguous(memory_format=torch.channels_last)
        if K64 != ktrue:  # re-pad the out channels the forward stored dense
            gy = F.pad(gy, (0, 0, 0, 0, 0, K64 - ktrue)).contiguous(
                memory_format=torch.channels_last)
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = _ext().conv2d_bwd_data(gy, wp, stride, padding, H, W,
                                        dil=dil, ctrue=-1)
        if ctx.needs_input_grad[1]:
            gw = _ext().conv2d_bwd_weight_prepadded(
                gy, xp, stride, wp.shape[2], wp.shape[3], dil=dil)
        return gx, gw, None, None, None, None


def _chanpad_nhwc(x, c_to):
    if x.shape[1] != c_to:
        x = F.pad(x, (0, 0, 0, 0, 0, c_to - x.shape[1]))
    return x.contiguous(memory_format=torch.channels_last)


def _prep_gen_weight(weight):
    """fp32 [K][C][R][S] -> bf16 channels_last, channel-padded to
    [pad64(K)][pad8(C)][R][S] (autograd slices the grads back through
    F.pad / .to)."""
    K, C, R, S = weight.shape
    C8 = _pad_mult(C, 8)
    K64 = _pad_mult(K, 64)
    wb = weight.to(torch.bfloat16)
    if (K64, C8) != (K, C):
        wb = F.pad(wb, (0, 0, 0, 0, 0, C8 - C, 0, K64 - K))
    return wb.contiguous(memory_format=torch.channels_last)


def _gen_conv(x, weight, bias, stride, padding, dil, wb=None):
    """bf16 NHWC generic conv via _GenConvFn, channel-padding both sides."""
    K, C = weight.shape[0], weight.shape[1]
    xb = _chanpad_nhwc(x, _pad_mult(C, 8))
    if wb is None:
        wb = _prep_gen_weight(weight)
    y = _GenConvFn.apply(xb, wb, stride, padding, dil, K)
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    return y


def _gen_conv_supported(mod, x):
    if not _native(x):
        return False
    k = mod.kernel_size
    d = mod.dilation
    s = mod.stride
    # any square filter qualifies: the MFMA kernel zero-fills the Kg tail
    # when R*S*pad8(C) % 64 != 0 (Net 5x5, Net1/Net2 valid-3x3 shapes)
    return not (k[0] != k[1] or d[0] != d[1] or s[0] != s[1]
                or s[0] not in (1, 2) or mod.groups != 1)


class FedConvGeneric(nn.Conv2d):
    """Conv2d routed through the generic MFMA path when the shape allows
    (R*S*pad8(C) % 64 == 0 — true for all VAE/CPC convs); stock F.conv2d
    otherwise and on CPU."""

    def forward(self, x):
        if _gen_conv_supported(self, x) and self.padding[0] == self.padding[1]:
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if x.dtype == torch.bfloat16:
                wb = _cached_frozen(self, "_wgen", self.weight,
                                    lambda: _prep_gen_weight(self.weight))
                return _gen_conv(x, self.weight, self.bias, self.stride[0],
                                 self.padding[0], self.dilation[0], wb=wb)
        return F.conv2d(x, self.weight.to(x.dtype),
                        self.bias.to(x.dtype) if self.bias is not None else None,
                        self.stride, self.padding, self.dilation, self.groups)


class _DilatePadFn(torch.autograd.Function):
    """y[n][c][h*str+pt][w*str+pl] = x: the input-dilation step of a
    transposed conv, on the vectorized NHWC pad/dilate kernel; backward is
    a strided slice."""

    @staticmethod
    def forward(ctx, x, str_, pt, pb, pl, pr):
        ctx.meta = (str_, pt, pl, x.shape[2], x.shape[3])
        return _ext().dilate_pad(x, pt, pb, pl, pr, str_)

    @staticmethod
    def backward(ctx, gy):
        str_, pt, pl, H, W = ctx.meta
        gx = gy[:, :, pt:pt + (H - 1) * str_ + 1:str_,
                pl:pl + (W - 1) * str_ + 1:str_]
        return gx.contiguous(memory_format=torch.channels_last), \
            None, None, None, None, None


class FedConvTranspose2d(nn.ConvTranspose2d):
    """ConvTranspose2d as dilate-input + stride-1 MFMA conv with the
    spatially-flipped, in/out-swapped weight (the exact adjoint of the
    forward conv; same identity the ResNet bwd-data path uses).  Covers the
    VAE decoder 4x4-stride-2 deconvs (simple_models.py:262-265, 336-340)."""

    def forward(self, x, output_size=None):
        k = self.kernel_size
        s = self.stride
        p = self.padding
        op = self.output_padding
        C8 = _pad_mult(self.in_channels, 8)
        ok = (output_size is None and k[0] == k[1] and s[0] == s[1]
              and p[0] == p[1] and self.dilation == (1, 1)
              and self.groups == 1 and k[0] - 1 - p[0] >= 0 and _native(x))
        if ok:
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            ok = x.dtype == torch.bfloat16
        if not ok:
            return super().forward(x, output_size)
        # weight [Cin][Cout][R][S] -> conv weight [Cout][Cin][R][S], rot180
        wr = self.weight.flip(2, 3).transpose(0, 1)
        wb = _cached_frozen(self, "_wgen", self.weight,
                            lambda: _prep_gen_weight(wr))
        pt = k[0] - 1 - p[0]
        xb = _chanpad_nhwc(x, C8)
        xd = _DilatePadFn.apply(xb, s[0], pt, pt + op[0], pt, pt + op[1])
        yb = _gen_conv(xd, wr, self.bias, 1, 0, 1, wb=wb)
        return yb


class _DilatedBankFn(torch.autograd.Function):
    """Fused multi-dilation conv bank forward (one launch for all taps via
    a block-diagonal combined weight, csrc conv2d_dilated_bank); backward
    decomposes per tap onto the existing bwd-data / bwd-weight kernels.
    Forward fusion is the hot win: CPC trains with LBFGS line searches,
    whose probes are forward-only (SURVEY §2a dilated-bank row)."""

    @staticmethod
    def forward(ctx, x, w2d, dils, pads, stride, R, ktrue, *ws):
        y = _ext().conv2d_dilated_bank(x, w2d, list(dils), list(pads),
                                       stride, R, ktrue)
        ctx.save_for_backward(x, *ws)
        ctx.meta = (dils, pads, stride, R)
        return y

    @staticmethod
    def backward(ctx, gy):
        x = ctx.saved_tensors[0]
        ws = ctx.saved_tensors[1:]
        dils, pads, stride, R = ctx.meta
        ext = _ext()
        H, W = x.shape[2], x.shape[3]
        gy = gy.contiguous(memory_format=torch.channels_last)
        gx = None
        gws = []
        kt = ws[0].shape[0]
        for t, wt in enumerate(ws):
            gy_t = gy[:, t * kt:(t + 1) * kt].contiguous(
                memory_format=torch.channels_last)
            if ctx.needs_input_grad[0]:
                wtb = wt.to(torch.bfloat16).contiguous(
                    memory_format=torch.channels_last)
                d = ext.conv2d_bwd_data(gy_t, wtb, stride, pads[t], H, W,
                                        dil=dils[t])
                gx = d if gx is None else gx + d
            if ctx.needs_input_grad[7 + t]:
                gws.append(ext.conv2d_bwd_weight(
                    gy_t, x, stride, pads[t], R, R, dil=dils[t]))
            else:
                gws.append(None)
        return (gx, None, None, None, None, None, None, *gws)


def dilated_bank(x, mods):
    """elu-less fused bank: x through every conv in `mods` (shared input,
    per-module dilation/padding), outputs channel-concatenated.  Equivalent
    to torch.cat([m(x) for m in mods], dim=1)."""
    dils = [m.dilation[0] for m in mods]
    pads = [m.padding[0] for m in mods]
    stride = mods[0].stride[0]
    R = mods[0].kernel_size[0]
    C8 = _pad_mult(mods[0].in_channels, 8)
    kt = mods[0].out_channels
    ktrue = kt * len(mods)
    xb = _chanpad_nhwc(x, C8)

    m0 = mods[0]
    vers = tuple(m.weight._version for m in mods)
    cache = getattr(m0, "_wbank", None)
    frozen = not any(m.weight.requires_grad for m in mods)
    if frozen and cache is not None and cache[0] == vers:
        w2d = cache[1]
    else:
        kg = R * R * C8
        with torch.set_grad_enabled(False):
            w2d = x.new_zeros(64, kg * len(mods), dtype=torch.bfloat16)
            for t, m in enumerate(mods):
                wt = m.weight.detach().to(torch.bfloat16)
                if wt.shape[1] != C8:
                    wt = F.pad(wt, (0, 0, 0, 0, 0, C8 - wt.shape[1]))
                w2d[t * kt:(t + 1) * kt, t * kg:(t + 1) * kg] = \
                    wt.permute(0, 2, 3, 1).reshape(kt, kg)
        if frozen:
            m0._wbank = (vers, w2d)
    y = _DilatedBankFn.apply(xb, w2d, dils, pads, stride, R, ktrue,
                             *[m.weight for m in mods])
    bias = torch.cat([m.bias for m in mods]) if mods[0].bias is not None \
        else None
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    return y


class FedConv2d(nn.Conv2d):
    def forward(self, x):
        # producer-padded input? (bn_elu(pad_out=...) fusion: x is the
        # interior view, mk[0] the padded buffer.)  A marker mismatch is
        # harmless — x alone is fully valid, the marker is just skipped.
        mk = getattr(x, "_fedkit_padded", None)
        pbuf = mk[0] if (mk is not None and mk[1] == self.padding[0]
                         and self.padding[0] > 0) else None
        if _native(x) and self.bias is None and self.groups == 1 \
                and self.dilation == (1, 1) and self.kernel_size[0] in (1, 3):
            if x.dtype != torch.bfloat16 and torch.is_autocast_enabled():
                x = x.to(torch.bfloat16)
            if x.dtype == torch.bfloat16:
                # the MI355X fast path: bf16 NHWC through the MFMA kernels;
                # the fp32->bf16 weight cast is autograd-tracked, so bwd-weight
                # gradients land on the fp32 master copy.  With a producer-
                # padded buffer, x is only the autograd edge — do NOT
                # materialize the (non-contiguous) interior view.
                if pbuf is None:
                    x = x.contiguous(memory_format=torch.channels_last)
                once = self.__dict__.pop("_w16_once", None)
                if once is not None and once[0] == self.weight._version:
                    w = once[1]
                else:
                    w = _cached_frozen(
                        self, "_w16", self.weight,
                        lambda: self.weight.to(torch.bfloat16).contiguous(
                            memory_format=torch.channels_last))
                y, part = _ConvFn.apply(x, w, self.stride[0],
                                        self.padding[0], pbuf)
                if self.training and part.numel():
                    y._fedkit_bn_stats = part
                return y
        return F.conv2d(x, self.weight.to(x.dtype),
                        self.bias.to(x.dtype) if self.bias is not None else None,
                        self.stride, self.padding, self.dilation, self.groups)
