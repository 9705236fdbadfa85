"""Fused conv-bias epilogues (bias+relu, bias+residual-add) for the Atari
ResNet hot path.

MIOpen applies conv bias as a separate full-tensor kernel and torch's
relu / add are two more HBM passes; running the convs bias-free and folding
the bias into the next elementwise op removes ~2 passes per residual block
(hip/kernels.hip bias_* kernels). The section conv's bias commutes with the
per-channel maxpool and is carried as a "pending" bias into the next block
(see models/atari.py).

Math is identical to the reference model (examples/atari/models.py:9-153);
only kernel fusion differs. CPU / no-kernel paths keep plain torch ops.
"""
import os

import torch

_kernels = None


def _k():
    global _kernels
    if _kernels is None:
        from moolib_amd import _kernels as k  # loud ImportError on GPU boxes

        _kernels = k
    return _kernels


def available(x, out_channels=None):
    """Fused path wants: CUDA + kernels built + fused channel count %8, <=64
    (LDS bias accumulator). Checks `out_channels` when given (the conv that
    will feed the fused ops), else x's own channels."""
    if not x.is_cuda or os.environ.get("MOOLIB_AMD_NO_FUSED_BIAS"):
        return False
    c = out_channels if out_channels is not None else x.size(1)
    if c % 8 != 0 or c > 64:
        return False
    try:
        _k()
    except ImportError:
        return False
    return True


class _BiasRelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, b):
        y = _k().bias_relu_fwd(x.contiguous(memory_format=torch.channels_last), b)
        ctx.save_for_backward(y)
        ctx.b_dtype = b.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dx, db = _k().bias_relu_bwd(dy, y)
        return dx, db.to(ctx.b_dtype)


class _BiasAdd2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, b1, s, b2):
        y = _k().bias_add2_fwd(
            x.contiguous(memory_format=torch.channels_last),
            b1,
            s.contiguous(memory_format=torch.channels_last),
            b2,
        )
        ctx.has_b2 = b2 is not None
        ctx.b_dtype = b1.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        if dy.is_cuda and dy.is_contiguous(memory_format=torch.channels_last):
            # custom small-grid NHWC channel sum: aten's bf16 reduce runs
            # this at ~0.4 TB/s (see hip/kernels.hip channel_sum_kernel)
            db = _k().channel_sum_fp32(dy).to(ctx.b_dtype)
        else:
            db = dy.sum(dim=(0, 2, 3), dtype=torch.float32).to(ctx.b_dtype)
        return dy, db, dy, (db if ctx.has_b2 else None)


def bias_relu(x, b):
    """relu(x + b[c]) in one NHWC pass."""
    return _BiasRelu.apply(x, b.to(x.dtype))


def bias_add2(x, b1, s, b2=None):
    """x + b1[c] + s (+ b2[c]) in one NHWC pass (residual close)."""
    return _BiasAdd2.apply(x, b1.to(x.dtype), s, None if b2 is None else b2.to(x.dtype))
