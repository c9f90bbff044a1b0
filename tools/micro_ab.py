#!/usr/bin/env python3
"""A/B microbench for the env-gated conv paths (vpad bwd-data, packed-Q dw).

The gates are read once per process (static), so run this twice:
    python tools/micro_ab.py                      # new paths
    FEDKIT_NO_VPAD=1 python tools/micro_ab.py   # old paths (FEDKIT_QP=1 enables packed-Q dw)
Prints one line per shape: op, shape, mean us.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import fedkit.ops

ext = fedkit.ops.ext()
BATCH = 128


def timeit(fn, iters=100, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000.0   # us


def main():
    torch.manual_seed(0)
    tag = "OLD" if os.environ.get("FEDKIT_NO_VPAD") else "NEW"
    # bwd-data stride-1 shapes (vpad)
    for C, H, K in [(64, 32, 64), (128, 16, 128), (256, 8, 256),
                    (512, 4, 512)]:
        gy = torch.randn(BATCH, K, H, H, device="cuda",
                         dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        w = (torch.randn(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
             * 0.05).contiguous(memory_format=torch.channels_last)
        us = timeit(lambda: ext.conv2d_bwd_data(gy, w, 1, 1, H, H))
        print(f"{tag} bwd_data_s1 C{C} H{H} K{K}: {us:8.1f} us")
    # bwd-data stride-2 shapes (vpad v2: parity-class virtual dilate)
    for C, H, K in [(64, 32, 128), (128, 16, 256), (256, 8, 512)]:
        P = H // 2
        gy = torch.randn(BATCH, K, P, P, device="cuda",
                         dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        w = (torch.randn(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
             * 0.05).contiguous(memory_format=torch.channels_last)
        us = timeit(lambda: ext.conv2d_bwd_data(gy, w, 2, 1, H, H))
        print(f"{tag} bwd_data_s2 C{C} H{H} K{K}: {us:8.1f} us")
    # linear bwd-weight (the 126 us head case + the big Net1 fc1)
    for M, K, N in [(128, 512, 10), (128, 1600, 512)]:
        gy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        us = timeit(lambda: ext.linear_bwd_weight(gy, x, True))
        print(f"{tag} linear_dw M{M} K{K} N{N}: {us:8.1f} us")
    # layer4-class dw shapes (packed-Q)
    for C, H, K, stride in [(512, 4, 512, 1), (256, 8, 512, 2)]:
        P = H // stride
        x = torch.randn(BATCH, C, H, H, device="cuda",
                        dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        gy = torch.randn(BATCH, K, P, P, device="cuda",
                         dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        us = timeit(lambda: ext.conv2d_bwd_weight(gy, x, stride, 1, 3, 3))
        print(f"{tag} dw C{C} H{H} K{K} s{stride}: {us:8.1f} us")


if __name__ == "__main__":
    main()


def fwd_sweep():
    """Forward conv timings per flagship shape (run under env combos:
    FEDKIT_CONV_STAGES=2, FEDKIT_CONV_BM64=1)."""
    tag = []
    if os.environ.get("FEDKIT_CONV_BM64") == "1":
        tag.append("bm64")
    if os.environ.get("FEDKIT_CONV_STAGES") == "2":
        tag.append("st2")
    tag = "+".join(tag) or "default"
    for C, H, K, stride in [(64, 32, 64, 1), (128, 16, 128, 1),
                            (256, 8, 256, 1), (512, 4, 512, 1),
                            (64, 32, 128, 2), (128, 16, 256, 2),
                            (256, 8, 512, 2)]:
        x = torch.randn(BATCH, C, H, H, device="cuda",
                        dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last)
        w = (torch.randn(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
             * 0.05).contiguous(memory_format=torch.channels_last)
        xp = ext.conv2d_pad_input(x, 1)
        us = timeit(lambda: ext.conv2d_fwd_prepadded(xp, w, stride))
        print(f"FWD[{tag}] C{C} H{H} K{K} s{stride}: {us:8.1f} us")


if os.environ.get("FEDKIT_FWD_SWEEP") == "1":
    fwd_sweep()
