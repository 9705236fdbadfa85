"""Fused masked-LSTM sequence scan (gfx950 MFMA kernel dispatch).

Replaces the T-step Python loop over nn.LSTM in the IMPALA model with one
kernel launch per direction (see hip/lstm.hip.inc). The input projection
x @ W_ih^T + b runs as one hipBLASLt GEMM over all T steps; the recurrent
part (h @ W_hh^T on mfma_f32_16x16x32_bf16, gates, done-masking) runs in
LDS-resident state. Hidden size fixed at 256 (the IMPALA config).
"""
import os

import torch


def _kernels():
    try:
        from moolib_amd import _kernels as k

        return k
    except ImportError:
        return None


def available(hidden_size, device):
    return (
        hidden_size == 256
        and device.type == "cuda"
        and _kernels() is not None
        and not os.environ.get("MOOLIB_AMD_NO_LSTM_KERNEL")
    )


def pack_mfma_b(w):
    """Pack W [K, N] into per-lane mfma_f32_16x16x32_bf16 B-fragment order:
    fragment (ct, kk), lane l = hi*16+lo, elem e -> W[kk*32+hi*8+e][ct*16+lo],
    laid out so each lane loads its 8 bf16 with one 16-byte read."""
    K, N = w.shape
    v = w.view(K // 32, 4, 8, N // 16, 16)  # [kk, hi, e, ct, lo]
    return v.permute(3, 0, 1, 4, 2).contiguous().view(-1)


class _FusedLSTMScan(torch.autograd.Function):
    @staticmethod
    def forward(ctx, X, notdone, h0, c0, w_hh):
        k = _kernels()
        w = w_hh.detach().to(torch.bfloat16)
        packed_t = pack_mfma_b(w.t().contiguous())  # [H, 4H]: K=256, N=1024
        H, gates, cs, hT, cT = k.lstm_fused_fwd(
            X.contiguous(), notdone.contiguous(), h0.contiguous(), c0.contiguous(), packed_t
        )
        ctx.save_for_backward(gates, cs, c0, notdone, w, h0, H)
        return H, hT, cT

    @staticmethod
    def backward(ctx, dH, dhT, dcT):
        k = _kernels()
        gates, cs, c0, notdone, w, h0, H = ctx.saved_tensors
        packed_n = pack_mfma_b(w.contiguous())  # [4H, H]: K=1024, N=256
        empty = torch.Tensor()
        dG, dh0, dc0 = k.lstm_fused_bwd(
            gates,
            cs,
            c0,
            notdone,
            dH.contiguous().to(torch.bfloat16),
            dhT.contiguous().to(torch.bfloat16) if dhT is not None else empty,
            dcT.contiguous().float() if dcT is not None else empty,
            packed_n,
        )
        T, B, _ = dG.shape
        # dW_hh = dG^T @ h_masked_prev  (h_masked[t] = nd[t] * H[t-1], H[-1]=h0)
        Hm = torch.cat([h0.unsqueeze(0), H[:-1]]) * notdone.unsqueeze(-1).to(H.dtype)
        dW = dG.reshape(T * B, -1).t().float() @ Hm.reshape(T * B, -1).float()
        return dG, None, dh0, dc0, dW

    # note: returned grads match input dtypes: X bf16 <- dG bf16; h0 bf16 <- dh0 bf16;
    # c0 f32 <- dc0 f32; w_hh gets fp32 (autograd casts to the param dtype).


def fused_lstm_scan(X, notdone, h0, c0, w_hh):
    """X [T,B,1024] bf16; notdone [T,B] f32; h0 [B,256] bf16; c0 [B,256] f32.
    Returns (H [T,B,256] bf16, hT [B,256] bf16, cT [B,256] f32)."""
    return _FusedLSTMScan.apply(X, notdone, h0, c0, w_hh)
