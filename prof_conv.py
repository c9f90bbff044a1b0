#!/usr/bin/env python3
"""Tight loop over ONE conv shape for rocprofv3 PMC counter collection.

    rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
        SQ_ACTIVE_INST_ANY FETCH_SIZE -- python prof_conv.py --shape layer1

One kernel dominates the trace so per-kernel counter rows are unambiguous.
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

SHAPES = {
    # Cin, H, Kout, stride, ksize
    "layer1": (64, 32, 64, 1, 3),
    "layer2": (128, 16, 128, 1, 3),
    "layer3": (256, 8, 256, 1, 3),
    "layer4": (512, 4, 512, 1, 3),
    "down2": (64, 32, 128, 2, 3),
    "conv1": (3, 32, 64, 1, 3),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--shape", default="layer1", choices=list(SHAPES))
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--op", default="fwd", choices=["fwd", "bwd_data", "bwd_weight"])
    args = ap.parse_args()
    import fedkit.ops
    ext = fedkit.ops.require_ext()
    Cin, H, Kout, stride, ks = SHAPES[args.shape]
    pad = 1
    x = torch.randn(128, Cin, H, H, device="cuda", dtype=torch.bfloat16
                    ).contiguous(memory_format=torch.channels_last)
    w = torch.randn(Kout, Cin, ks, ks, device="cuda", dtype=torch.bfloat16
                    ).contiguous(memory_format=torch.channels_last)
    P = (H + 2 * pad - ks) // stride + 1
    gy = torch.randn(128, Kout, P, P, device="cuda", dtype=torch.bfloat16
                     ).contiguous(memory_format=torch.channels_last)
    torch.cuda.synchronize()
    for _ in range(args.iters):
        if args.op == "fwd":
            ext.conv2d_fwd(x, w, stride, pad)
        elif args.op == "bwd_data":
            ext.conv2d_bwd_data(gy, w, stride, pad, H, H)
        else:
            ext.conv2d_bwd_weight(gy, x, stride, pad, ks, ks)
    torch.cuda.synchronize()
    print("done", args.shape, args.op)


if __name__ == "__main__":
    main()
