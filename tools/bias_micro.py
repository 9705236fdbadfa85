#!/usr/bin/env python
"""Micro-timing for the fused bias elementwise kernels at learner shapes.

bias_relu_bwd runs at ~3.4 TB/s blended (profiles/r4b_final_kernel_stats):
this times it in isolation per shape so block-cap sweeps
(MOOLIB_AMD_BIAS_BWD_BLOCKS) resolve differences bench-level noise hides.

Run on an MI355X:  python tools/bias_micro.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from moolib_amd import _kernels

SHAPES = [(672, 16, 42, 42), (672, 32, 21, 21), (672, 32, 11, 11)]


def time_op(fn, iters=200):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1000.0 / iters  # us


def main():
    cap = os.environ.get("MOOLIB_AMD_BIAS_BWD_BLOCKS", "2048")
    for shape in SHAPES:
        n, c, h, w = shape
        y = torch.randn(shape, device="cuda", dtype=torch.bfloat16).to(
            memory_format=torch.channels_last
        )
        dy = torch.randn_like(y)
        us = time_op(lambda: _kernels.bias_relu_bwd(dy, y))
        bytes_moved = y.numel() * 2 * 3  # dy + y read, dx write
        print(
            "bias_relu_bwd cap=%s %-22s %7.2f us  %.2f TB/s"
            % (cap, str(shape), us, bytes_moved / us / 1e6)
        )


if __name__ == "__main__":
    main()
