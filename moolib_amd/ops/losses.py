"""IMPALA losses: policy gradient, baseline, entropy.

Capability parity with the reference's loss helpers
(examples/vtrace/experiment.py:62-82); our own implementation. On GPU the
fused HIP kernel (moolib_amd._kernels.impala_loss_fwd/bwd) computes all
three losses and the logits/baseline gradients in one pass over [T*B, A]
(see moolib_amd.ops.fused_loss); these plain-torch forms are the CPU path
and the numerics reference.
"""
import torch
import torch.nn.functional as F


def policy_gradient_loss(logits, actions, advantages):
    """-E[log pi(a|x) * adv], mean over T*B. logits [T,B,A], actions [T,B]."""
    logp = F.log_softmax(logits, dim=-1)
    chosen = torch.gather(logp, -1, actions.unsqueeze(-1).to(torch.int64)).squeeze(-1)
    return torch.mean(-chosen * advantages.detach())


def baseline_loss(advantages):
    """0.5 * mean((vs - baseline)^2)."""
    return 0.5 * torch.mean(advantages**2)


def entropy_loss(logits):
    """-mean entropy (to be *added* with a positive cost, matching reference)."""
    policy = F.softmax(logits, dim=-1)
    log_policy = F.log_softmax(logits, dim=-1)
    entropy_per_step = torch.sum(-policy * log_policy, dim=-1)
    return -torch.mean(entropy_per_step)
