#!/usr/bin/env python
"""Microbenchmark: conv3x3 MFMA kernel vs MIOpen at IMPALA shapes.

Per shape, times ours (each row-tile RT variant + the auto pick) and
F.conv2d (MIOpen, after its find settles), with relu fused in ours and
counted for MIOpen. Run on an MI355X:
    python tools/conv3x3_micro.py > profiles/rXX_conv3x3_micro.txt
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from moolib_amd.ops import conv3x3 as c3


def timeit(fn, iters=50):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    torch.manual_seed(0)
    shapes = [
        # (N, C, H, W, K) — actor path then learner-sized M
        (128, 16, 42, 42, 16),
        (128, 16, 42, 42, 32),
        (128, 32, 21, 21, 32),
        (128, 32, 11, 11, 32),
        (672, 16, 42, 42, 16),
        (672, 32, 21, 21, 32),
    ]
    print(f"{'shape':<18} {'C->K':>6} {'M':>8} {'rt1':>7} {'rt2':>7} {'rt4':>7} {'auto':>7} {'miopen':>8}  ratio(auto/miopen)")
    for N, C, H, W, K in shapes:
        x = torch.randn(N, C, H, W, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        w = torch.randn(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
        wp = c3.pack_weight(w)
        b = torch.randn(K, device="cuda", dtype=torch.bfloat16)

        times = {}
        for rt in (1, 2, 4, 0):
            times[rt] = timeit(
                lambda rt=rt: c3.conv3x3(x, wp, K, relu_in=True, epi=c3.EPI_BIAS, bias1=b, rt=rt)
            )
        wcl = w.to(memory_format=torch.channels_last)
        t_mi = timeit(lambda: F.conv2d(F.relu(x), wcl, b, padding=1))
        M = N * H * W
        print(
            f"[{N},{C},{H},{W}]".ljust(18)
            + f" {C}->{K}".rjust(6)
            + f" {M:>8}"
            + "".join(f" {times[rt]:>7.3f}" for rt in (1, 2, 4, 0))
            + f" {t_mi:>8.3f}  {times[0] / t_mi:.2f}"
        )


def wgrad_micro():
    from moolib_amd import _kernels

    print("\nwgrad (dW) micro:")
    print(f"{'shape':<18} {'C->K':>6} {'ours_ms':>8} {'miopen_ms':>10}  ratio")
    for N, C, H, W, K in [
        (672, 16, 42, 42, 16),
        (672, 16, 42, 42, 32),
        (672, 32, 21, 21, 32),
        (672, 32, 11, 11, 32),
    ]:
        x = torch.randn(N, C, H, W, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        dy = torch.randn(N, K, H, W, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        w = torch.zeros(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
        t_ours = timeit(lambda: _kernels.wgrad3x3_nhwc(x, dy))
        t_mi = timeit(
            lambda: torch.ops.aten.convolution_backward(
                dy, x, w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                [False, True, False],
            )[1]
        )
        print(f"[{N},{C},{H},{W}]".ljust(18) + f" {C}->{K}".rjust(6)
              + f" {t_ours:>8.3f} {t_mi:>10.3f}  {t_ours/t_mi:.2f}")


if __name__ == "__main__":
    main()
    wgrad_micro()
