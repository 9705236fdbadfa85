"""Fused MFMA 3x3 NHWC conv (actor inference AND learner autograd paths).

Wraps hip/conv3x3.hip.inc: implicit-GEMM conv on mfma_f32_16x16x32_bf16
with the producer relu (+pending bias) fused into the input load and the
bias / bias+relu / bias+residual epilogue fused into the store. An IMPALA
residual block (reference examples/atari/models.py:30-42) becomes two
kernel launches on the actor. The learner uses `conv3x3_autograd`
(forward + input-gradient on the same kernel; weight gradient via MIOpen
wrw by default, our wgrad3x3 kernel opt-in) and `conv1_u8_autograd` for
the uint8 first layer.
"""
import os

import torch

from moolib_amd.ops.lstm import pack_mfma_b

EPI_NONE, EPI_BIAS, EPI_BIAS_RELU, EPI_BIAS_ADD = 0, 1, 2, 3

_kernels = None


def _k():
    global _kernels
    if _kernels is None:
        from moolib_amd import _kernels as k

        _kernels = k
    return _kernels


def available(C, K):
    # Default ON for the actor path since r2: with packed weights cached in
    # version-checked persistent buffers (repack() once per optimizer step;
    # the r1 regression was per-replay repacking inside the captured
    # forward), same-box interleaved A/B measured the kernel path ~4%
    # faster end-to-end (profiles/evidence/r2_{on,off}*.json: 60.7k vs 58.4k and
    # 58.7k vs 56.2k f/s). MOOLIB_AMD_CONV3_KERNEL=0 reverts to MIOpen.
    if os.environ.get("MOOLIB_AMD_CONV3_KERNEL", "1") == "0":
        return False
    if C not in (16, 32) or K not in (16, 32):
        return False
    try:
        _k()
    except ImportError:
        return False
    return True


def pack_weight(w):
    """nn.Conv2d weight [K, C, 3, 3] -> packed MFMA B fragments.

    GEMM B is W[kd][kout] with kd = (kh*3+kw)*C + c, zero-padded so the
    Kdim is a multiple of 32, then laid out in pack_mfma_b fragment order.
    """
    K, C, kh, kw = w.shape
    assert kh == 3 and kw == 3
    gem = w.permute(2, 3, 1, 0).reshape(9 * C, K)
    kk32 = (9 * C + 31) // 32 * 32
    if kk32 != 9 * C:
        gem = torch.cat([gem, gem.new_zeros(kk32 - 9 * C, K)])
    return pack_mfma_b(gem.to(torch.bfloat16).contiguous())


def packed_buffer(conv):
    """Version-checked persistent packed-weight buffer for an nn.Conv2d.

    Packing costs ~6 small kernels per conv; done inline it dominates the
    graphed actor forward (measured: the kernel itself BEATS MIOpen at
    actor shapes in isolation — profiles/r01_conv3x3_micro.txt — but
    per-replay repacking erased the win). The buffer has a stable address,
    so hipGraphs may capture reads of it; refresh after weight updates
    with repack() (graph-capturable: pack + copy_ into the buffer).
    """
    w = conv.weight
    ver = w._version
    cache = getattr(conv, "_c3_cache", None)
    if cache is None:
        conv._c3_cache = cache = [ver, pack_weight(w.detach())]
        return cache[1]
    if cache[0] != ver:
        cache[1].copy_(pack_weight(w.detach()))
        cache[0] = ver
    return cache[1]


def repack(module):
    """Refresh every cached packed weight in `module` (call after the
    optimizer step; safe inside a hipGraph capture).

    On GPU with bf16 weights this is ONE kernel launch for all convs and
    both directions (repack3x3_batched); the torch-op path (flip/permute/
    cat per conv, ~14 kernels each) remains the fallback."""
    import torch.nn as nn

    ws, fs, ds, cps, caches = [], [], [], [], []
    fallback = []
    for m in module.modules():
        if isinstance(m, nn.Conv2d):
            cache = getattr(m, "_c3_cache", None)
            dcache = getattr(m, "_c3_dgrad_cache", None)
            c1cache = getattr(m, "_c1_pack_cache", None)
            if cache is None and dcache is None and c1cache is None:
                continue
            w = m.weight.detach()
            if w.is_cuda and w.dtype == torch.bfloat16:
                if cache is not None or dcache is not None:
                    ws.append(w)
                    fs.append(cache[1] if cache is not None else w.new_empty(0))
                    ds.append(dcache[1] if dcache is not None else w.new_empty(0))
                    cps.append(0)
                    caches.append((m, cache, dcache))
                if c1cache is not None:
                    # the C=8-padded first-conv pack rides the same launch
                    ws.append(w)
                    fs.append(c1cache[1])
                    ds.append(w.new_empty(0))
                    cps.append(8)
                    caches.append((m, c1cache, None))
            else:
                fallback.append((m, cache, dcache, c1cache))
    if ws:
        _k().repack3x3_batched(ws, fs, ds, cps)
        for m, cache, dcache in caches:
            if cache is not None:
                cache[0] = m.weight._version
            if dcache is not None:
                dcache[0] = m.weight._version
    for m, cache, dcache, c1cache in fallback:
        if cache is not None:
            cache[1].copy_(pack_weight(m.weight.detach()))
            cache[0] = m.weight._version
        if dcache is not None:
            dcache[1].copy_(pack_weight_dgrad(m.weight.detach()))
            dcache[0] = m.weight._version
        if c1cache is not None:
            import torch.nn.functional as F

            w8 = F.pad(m.weight.detach(), (0, 0, 0, 0, 0, 8 - m.weight.shape[1]))
            c1cache[1].copy_(pack_weight(w8))
            c1cache[0] = m.weight._version


def pack_weight_dgrad(w):
    """Packed fragments for the INPUT-gradient conv.

    For stride-1 pad-1 3x3, dx = conv3x3_s1_p1(dy, W') with
    W'[c,k,kh,kw] = W[k,c,2-kh,2-kw] (transpose + 180-degree tap rotation),
    so the backward data pass reuses the forward kernel unchanged."""
    wd = w.flip(2, 3).permute(1, 0, 2, 3).contiguous()
    return pack_weight(wd)


def dgrad_buffer(conv):
    """Version-checked persistent dgrad-packed buffer (see packed_buffer)."""
    w = conv.weight
    ver = w._version
    cache = getattr(conv, "_c3_dgrad_cache", None)
    if cache is None:
        conv._c3_dgrad_cache = cache = [ver, pack_weight_dgrad(w.detach())]
        return cache[1]
    if cache[0] != ver:
        cache[1].copy_(pack_weight_dgrad(w.detach()))
        cache[0] = ver
    return cache[1]


class _Conv3x3Fn(torch.autograd.Function):
    """3x3/s1/p1 NHWC conv with fwd AND dgrad on the MFMA kernel.

    Weight gradient defaults to aten.convolution_backward (MIOpen wrw
    igemm); MOOLIB_AMD_WGRAD_KERNEL=1 switches to our wgrad3x3 kernel.
    Drop-in for F.conv2d(x, w, None, padding=1) on supported shapes."""

    @staticmethod
    def forward(ctx, x, weight, conv_module):
        xb = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        y = conv3x3(xb, packed_buffer(conv_module), weight.shape[0])
        ctx.save_for_backward(xb, weight)
        ctx.conv_module = conv_module
        ctx.x_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        m = ctx.conv_module
        dy = dy.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = conv3x3(dy, dgrad_buffer(m), w.shape[1])
            if dx.dtype != ctx.x_dtype:
                dx = dx.to(ctx.x_dtype)
        if ctx.needs_input_grad[1]:
            if os.environ.get("MOOLIB_AMD_WGRAD_KERNEL"):
                # Opt-in: the row-slab wgrad kernel is numerics-correct
                # (TestWgradKernel) and, after the accumulator-scratch fix,
                # within 1.1-1.25x of MIOpen's igemm_wrw at IMPALA learner
                # shapes (102us vs 93us, 69us vs 56us —
                # profiles/evidence/r2l_sweep.txt). MIOpen keeps the default until
                # an A/B shows parity; the remaining gap is the per-slab
                # atomic fold.
                K, C = w.shape[0], w.shape[1]
                # [9C-padded, K] fp32 GEMM gradient -> [K, C, 3, 3]
                g = _k().wgrad3x3_nhwc(x, dy)[: 9 * C]
                dw = g.view(3, 3, C, K).permute(3, 2, 0, 1)
            else:
                # wrw requires x/dy/w in one dtype; under plain autocast the
                # master weights are fp32 while the saved x and dy are bf16
                wb = w if w.dtype == dy.dtype else w.to(dy.dtype)
                dw = torch.ops.aten.convolution_backward(
                    dy, x, wb, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                    [False, True, False],
                )[1]
            if dw.dtype != w.dtype:  # bf16 weights (shadow mode) / fp32 master
                dw = dw.to(w.dtype)
        return dx, dw, None


def conv3x3_autograd(x, conv_module):
    """Autograd-capable conv through the MFMA kernel (learner path)."""
    return _Conv3x3Fn.apply(x, conv_module.weight, conv_module)


def conv1_packed_buffer(conv):
    """C=8 zero-padded MFMA pack for the 4-channel first conv.

    The MFMA template's A fragment spans 8 contiguous channels of one
    tap, so C=4 runs as C=8 with zero channels (25% padded MACs beat the
    scalar direct conv by a wide margin at learner batch). Version-checked
    like packed_buffer; refreshed by repack() (the batched repack kernel
    zero-fills channels >= the real C)."""
    import torch.nn.functional as F

    w = conv.weight
    ver = w._version
    cache = getattr(conv, "_c1_pack_cache", None)
    if cache is None or cache[0] != ver:
        w8 = F.pad(w.detach(), (0, 0, 0, 0, 0, 8 - w.shape[1]))
        packed = pack_weight(w8)
        if cache is None:
            conv._c1_pack_cache = cache = [ver, packed]
        else:
            cache[1].copy_(packed)
            cache[0] = ver
    return cache[1]


class _Conv1U8Fn(torch.autograd.Function):
    """First conv (uint8 frames -> 16ch), learner path: frames decode to a
    C=8 zero-padded bf16 NHWC tensor (one fused kernel) and the conv runs
    on the MFMA template with the bias fused into the epilogue; dW comes
    from our wgrad kernel on the saved padded activation, db is a sum.
    No MIOpen anywhere in the layer (frames never need an input
    gradient). The scalar conv1_u8 kernel remains the actor (no-grad,
    hipGraph-captured) path where N is small."""

    @staticmethod
    def forward(ctx, x_u8, weight, bias, conv_module, scale):
        k = _k()
        xb8 = k.frames_u8_to_bf16_nhwc(x_u8, scale, 8)
        y = conv3x3(
            xb8, conv1_packed_buffer(conv_module), weight.shape[0],
            epi=EPI_BIAS, bias1=bias.detach(),
        )
        ctx.save_for_backward(xb8, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        xb8, w = ctx.saved_tensors
        k = _k()
        dyb = dy.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        K, C = w.shape[0], w.shape[1]
        g = k.wgrad3x3_nhwc(xb8, dyb)[: 9 * 8]
        dw = g.view(3, 3, 8, K)[:, :, :C].permute(3, 2, 0, 1).to(w.dtype)
        # small-grid NHWC channel sum (hip/kernels.hip channel_sum_kernel)
        db = k.channel_sum_fp32(dyb).to(w.dtype)
        return None, dw, db, None, None


def conv1_u8_autograd(x_u8, conv_module, scale):
    return _Conv1U8Fn.apply(
        x_u8, conv_module.weight, conv_module.bias, conv_module, scale
    )


def conv3x3(x, w_packed, k, relu_in=False, bias_in=None, epi=EPI_NONE,
            bias1=None, res=None, bias2=None, rt=0):
    """out = conv3x3_s1_p1( relu(x + bias_in) if relu_in else x ) then
    epilogue: none / +bias1 / relu(+bias1) / +bias1+res(+bias2).
    rt>0 overrides the row-tiles-per-wave pick (microbenchmarking)."""
    to_bf = lambda t: None if t is None else t.to(torch.bfloat16)
    return _k().conv3x3_nhwc_fused(
        x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last),
        w_packed,
        k,
        relu_in,
        to_bf(bias_in),
        epi,
        to_bf(bias1),
        None if res is None else to_bf(res).contiguous(memory_format=torch.channels_last),
        to_bf(bias2),
        rt,
    )
