"""Custom NHWC 3x3/stride-2 max pooling (the IMPALA ResNet's pools).

torch's channels_last maxpool backward scatters gradients with atomics
(~186 us per call on the bench shapes); the gfx950 kernel pair here stores
a per-output window index in the forward and gathers in the backward (each
input element checks its <=4 covering windows) — atomic-free and
bandwidth-bound. Falls back to F.max_pool2d off-GPU.
"""
import os

import torch
import torch.nn.functional as F


def _kernels():
    try:
        from moolib_amd import _kernels as k

        return k
    except ImportError:
        return None


class _MaxPool3x3s2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        k = _kernels()
        out, idx = k.maxpool3x3s2_fwd(x.contiguous(memory_format=torch.channels_last))
        ctx.save_for_backward(idx)
        ctx.in_hw = (x.shape[2], x.shape[3])
        return out

    @staticmethod
    def backward(ctx, gout):
        (idx,) = ctx.saved_tensors
        k = _kernels()
        return k.maxpool3x3s2_bwd(gout, idx, ctx.in_hw[0], ctx.in_hw[1])


def maxpool3x3s2(x):
    """max_pool2d(x, 3, stride=2, padding=1) — fused NHWC path on MI355X."""
    if (
        x.is_cuda
        and x.shape[1] % 8 == 0
        and _kernels() is not None
        and not os.environ.get("MOOLIB_AMD_NO_POOL_KERNEL")
    ):
        return _MaxPool3x3s2.apply(x)
    return F.max_pool2d(x, 3, stride=2, padding=1)
