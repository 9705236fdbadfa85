"""Fused IMPALA loss (policy-gradient + baseline + entropy) on gfx950.

One HIP kernel computes all three loss terms AND the analytic gradients
w.r.t. logits and baseline (moolib_amd._kernels.impala_loss); autograd sees
a single Function whose backward just scales the precomputed gradients.
Numerics match the eager torch path in moolib_amd.ops.losses (the GPU test
asserts this against a plain fp32 torch reference).
"""
import torch

from . import losses as eager


def _kernels():
    import os

    if os.environ.get("MOOLIB_AMD_NO_LOSS_KERNEL"):
        return None
    try:
        from moolib_amd import _kernels as k

        return k
    except ImportError:
        return None


class _FusedImpalaLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, baseline, actions, pg_advantages, vs, entropy_cost, baseline_cost):
        k = _kernels()
        parts, glogits, gbaseline = k.impala_loss(
            logits.contiguous(),
            actions.contiguous(),
            pg_advantages.contiguous(),
            vs.contiguous(),
            baseline.contiguous(),
            float(entropy_cost),
            float(baseline_cost),
            1.0,
        )
        ctx.save_for_backward(glogits, gbaseline)
        ctx.shape = baseline.shape
        return parts.sum()

    @staticmethod
    def backward(ctx, grad_out):
        glogits, gbaseline = ctx.saved_tensors
        return (
            glogits * grad_out,
            (gbaseline * grad_out).view(ctx.shape),
            None,
            None,
            None,
            None,
            None,
        )


def impala_total_loss(logits, baseline, actions, pg_advantages, vs, entropy_cost, baseline_cost):
    """total = pg + baseline_cost*baseline + entropy_cost*entropy.

    logits [T,B,A] fp32 (requires_grad), baseline [T,B] fp32
    (requires_grad), actions [T,B] int64, pg_advantages/vs [T,B] fp32
    (no grad). Uses the fused gfx950 kernel on GPU, eager torch elsewhere.
    """
    if logits.is_cuda and _kernels() is not None:
        return _FusedImpalaLoss.apply(
            logits, baseline, actions, pg_advantages, vs, entropy_cost, baseline_cost
        )
    pg = eager.policy_gradient_loss(logits, actions, pg_advantages)
    bl = baseline_cost * eager.baseline_loss(vs - baseline)
    en = entropy_cost * eager.entropy_loss(logits)
    return pg + bl + en
