"""FedBatchNorm2d — BatchNorm2d whose GPU path is a hand-written NHWC kernel.

Subclasses nn.BatchNorm2d (parameter order / state_dict parity with the
reference's nn.BatchNorm2d modules, simple_models.py:138-146).

GPU path (csrc/batchnorm.hip): two-kernel NHWC BatchNorm —
  (1) per-channel mean/var reduction over N*H*W (fp32 accumulation, one pass,
      channels on the contiguous innermost axis so lanes read coalesced);
  (2) normalize + scale/shift apply, writing y in the input dtype.
Training stats + running-stat update happen on device; backward is the
standard two-pass (reduce dy, dy*xhat; then apply), all fp32 math on bf16 data.

CPU path: stock F.batch_norm (the numerics reference for the GPU tests).
Stats and running buffers stay fp32 even when activations are bf16
(SURVEY.md §7 hard part 4).
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


class _BnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                training, momentum, eps, conv_part=None):
        y, save_mean, save_invstd = _ext().bn_fwd(
            x, weight, bias, running_mean, running_var,
            bool(training), float(momentum), float(eps),
            conv_part=conv_part)
        ctx.save_for_backward(x, weight, save_mean, save_invstd)
        return y

    @staticmethod
    def backward(ctx, gy):
        x, weight, save_mean, save_invstd = ctx.saved_tensors
        gy = gy.contiguous(memory_format=torch.channels_last)
        gx, gw, gb = _ext().bn_bwd(gy, x, weight, save_mean, save_invstd)
        return gx, gw, gb, None, None, None, None, None, None


class _BnActFn(torch.autograd.Function):
    """BatchNorm with fused residual-add + ELU epilogue (one HBM pass).

    Replaces the reference's `elu(bn(conv(x)) [+ shortcut])` patterns
    (simple_models.py:150-153) with a single apply kernel.  Backward:
    g = elu'(y) * gy (from the saved post-activation output), which is both
    the residual grad and the BN-output grad.

    pad_out > 0 (round 2): the apply ALSO writes the PADDED image the
    next conv consumes (borders zeroed in-kernel) — removes the separate
    pad launch plus a full read+write pass of y.  The AUTOGRAD output is
    the interior VIEW of that buffer (unpadded shape), so every gradient
    stays in the unpadded domain — the consuming conv reads the padded
    buffer out-of-band (the `_fedkit_padded` marker) but computes its
    input gradient at the ORIGINAL geometry (no wasted border-gradient
    work); only the bn_bwd kernels' reads of the saved y use the pad
    geometry.  A producer-padded RESIDUAL ships its buffer via res_buf
    (read at interior coords in-kernel) while the interior view carries
    the autograd edge.
    """

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                training, momentum, eps, residual, conv_part=None,
                pad_out=0, res_pad=0, res_buf=None):
        y, save_mean, save_invstd = _ext().bn_fwd(
            x, weight, bias, running_mean, running_var,
            bool(training), float(momentum), float(eps),
            residual=res_buf if res_buf is not None else residual,
            elu=True, conv_part=conv_part,
            pad_out=int(pad_out), res_pad=int(res_pad))
        ctx.save_for_backward(x, weight, save_mean, save_invstd, y)
        ctx.has_res = residual is not None
        ctx.pad_out = int(pad_out)
        if pad_out:
            p = int(pad_out)
            ypad = y
            yview = y[:, :, p:-p, p:-p]
        else:
            ypad = y.new_empty(0)
            yview = y
        ctx.mark_non_differentiable(ypad)
        return yview, ypad

    @staticmethod
    def backward(ctx, gy, _gypad):
        x, weight, save_mean, save_invstd, y = ctx.saved_tensors
        gy = gy.contiguous(memory_format=torch.channels_last)
        # elu' is fused into both bn_bwd kernels (recomputed from the saved
        # output y, read at padded coords when pad_out was used); the
        # residual grad g streams out of the apply kernel on demand
        out = _ext().bn_bwd(gy, x, weight, save_mean, save_invstd,
                            elu_y=y, want_g=ctx.has_res,
                            pad_in=ctx.pad_out)
        gres = out[3] if ctx.has_res else None
        return (out[0], out[1], out[2], None, None, None, None, None, gres,
                None, None, None, None)


class FedBatchNorm2d(nn.BatchNorm2d):
    def forward(self, x):
        if _native(x):
            self._prep(x)
            part = getattr(x, "_fedkit_bn_stats", None) \
                if self.training else None
            x = x.contiguous(memory_format=torch.channels_last)
            return _BnFn.apply(x, self.weight, self.bias,
                               self.running_mean, self.running_var,
                               self.training, self.momentum, self.eps, part)
        return super().forward(x)

    def _prep(self, x):
        self._check_input_dim(x)
        if self.training and self.track_running_stats \
                and self.num_batches_tracked is not None:
            if self.momentum is None:
                # cumulative-average mode actually consumes the counter
                self.num_batches_tracked.add_(1)
            else:
                # the counter is checkpoint-only state (momentum fixed at
                # 0.1 in every reference model) — a device add_ here costs
                # one kernel launch per BN layer per step (~76 us/step on
                # ResNet18); accumulate host-side, materialize on save
                self._nbt_pending = getattr(self, "_nbt_pending", 0) + 1

    def _flush_nbt(self):
        """Materialize host-accumulated step counts into the
        num_batches_tracked buffer.  INVARIANT: the buffer may lag by
        `_nbt_pending` steps between flush points; flushes happen on
        state_dict save, state_dict load (reset), and train()/eval()
        transitions — any code reading the buffer directly mid-epoch
        (cross-rank BN sync, external tooling) must call this first
        (ADVICE r1)."""
        pending = getattr(self, "_nbt_pending", 0)
        if pending and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(pending)
        self._nbt_pending = 0

    def train(self, mode: bool = True):
        # flush at mode transitions so eval-time readers see the true count
        if getattr(self, "_nbt_pending", 0):
            self._flush_nbt()
        return super().train(mode)

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kwargs):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kwargs)


def bn_elu(bn: FedBatchNorm2d, x: torch.Tensor,
           residual: torch.Tensor = None, pad_out: int = 0) -> torch.Tensor:
    """elu(bn(x) [+ residual]) — fused on GPU, composed on CPU.

    pad_out > 0: the GPU path returns the interior view of a padded
    side-buffer (shipped via the `_fedkit_padded` marker) so the consuming
    conv skips its pad pass; the CPU path ignores it (the conv pads
    itself)."""
    if _native(x):
        bn._prep(x)
        part = getattr(x, "_fedkit_bn_stats", None) if bn.training else None
        x = x.contiguous(memory_format=torch.channels_last)
        res_pad = 0
        res_buf = None
        if residual is not None:
            # a producer-padded residual ships its padded buffer for the
            # in-kernel interior read; the view keeps the autograd edge
            mk = getattr(residual, "_fedkit_padded", None)
            if mk is not None:
                res_buf, res_pad = mk
                if res_buf.dtype != x.dtype:
                    res_buf = res_buf.to(x.dtype)
            else:
                residual = residual.contiguous(
                    memory_format=torch.channels_last)
                if residual.dtype != x.dtype:
                    residual = residual.to(x.dtype)
        y, ypad = _BnActFn.apply(x, bn.weight, bn.bias,
                                 bn.running_mean, bn.running_var,
                                 bn.training, bn.momentum, bn.eps, residual,
                                 part, pad_out, res_pad, res_buf)
        if pad_out:
            y._fedkit_padded = (ypad.detach(), pad_out)
        return y
    y = nn.BatchNorm2d.forward(bn, x)
    if residual is not None:
        y = y + residual
    return F.elu(y)
