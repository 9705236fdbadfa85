"""V-trace off-policy correction (IMPALA, Espeholt et al. 2018).

Computes the v_s targets and policy-gradient advantages from behavior/target
policy logits. Capability parity with the reference's
examples/common/vtrace.py; implementation is our own, written from the
paper's recursion:

    delta_t = rho_t (r_t + gamma_t V(x_{t+1}) - V(x_t))
    v_s - V(x_s) = sum_{t>=s} gamma^{t-s} (prod_{i<t} c_i) delta_t
computed as the backward scan
    A_t = delta_t + gamma_t c_t A_{t+1}.

On an MI355X GPU the backward scan + advantage computation runs as one
fused HIP kernel (moolib_amd._kernels.vtrace); the T-step Python loop of
the reference launches ~6 kernels per timestep. CPU / fallback path is
plain torch and serves as the numerics reference for the kernel tests.
"""
import collections

import torch
import torch.nn.functional as F

VTraceFromLogitsReturns = collections.namedtuple(
    "VTraceFromLogitsReturns",
    [
        "vs",
        "pg_advantages",
        "log_rhos",
        "behavior_action_log_probs",
        "target_action_log_probs",
    ],
)

VTraceReturns = collections.namedtuple("VTraceReturns", "vs pg_advantages")


def action_log_probs(policy_logits, actions):
    """log pi(a_t | x_t) for int actions under softmax logits [T, B, A]."""
    logp = F.log_softmax(policy_logits, dim=-1)
    return torch.gather(logp, -1, actions.unsqueeze(-1).to(torch.int64)).squeeze(-1)


def _kernels():
    import os

    if os.environ.get("MOOLIB_AMD_NO_VTRACE_KERNEL"):
        return None
    try:
        from moolib_amd import _kernels as k

        return k
    except ImportError:
        return None


@torch.no_grad()
def from_importance_weights(
    log_rhos,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    """V-trace from log importance weights. All inputs [T, B] (float32),
    bootstrap_value [B]. Returns VTraceReturns(vs, pg_advantages)."""
    k = _kernels()
    if k is not None and log_rhos.is_cuda:
        vs, pg = k.vtrace_from_log_rhos(
            log_rhos.float().contiguous(),
            discounts.float().contiguous(),
            rewards.float().contiguous(),
            values.float().contiguous(),
            bootstrap_value.float().contiguous(),
            float(clip_rho_threshold if clip_rho_threshold is not None else -1.0),
            float(clip_pg_rho_threshold if clip_pg_rho_threshold is not None else -1.0),
        )
        return VTraceReturns(vs=vs, pg_advantages=pg)

    rhos = torch.exp(log_rhos)
    clipped_rhos = rhos.clamp(max=clip_rho_threshold) if clip_rho_threshold is not None else rhos
    cs = rhos.clamp(max=1.0)
    T = discounts.shape[0]

    next_values = torch.cat([values[1:], bootstrap_value.unsqueeze(0)], dim=0)
    deltas = clipped_rhos * (rewards + discounts * next_values - values)

    # Backward scan: A_t = delta_t + gamma_t c_t A_{t+1}
    acc = torch.zeros_like(bootstrap_value)
    out = torch.empty_like(deltas)
    for t in reversed(range(T)):
        acc = deltas[t] + discounts[t] * cs[t] * acc
        out[t] = acc
    vs = out + values

    next_vs = torch.cat([vs[1:], bootstrap_value.unsqueeze(0)], dim=0)
    clipped_pg_rhos = (
        rhos.clamp(max=clip_pg_rho_threshold) if clip_pg_rho_threshold is not None else rhos
    )
    pg_advantages = clipped_pg_rhos * (rewards + discounts * next_vs - values)
    return VTraceReturns(vs=vs, pg_advantages=pg_advantages)


def from_logits(
    behavior_policy_logits,
    target_policy_logits,
    actions,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    """V-trace for softmax policies; inputs [T, B, A] logits, [T, B] rest."""
    target_lp = action_log_probs(target_policy_logits, actions)
    behavior_lp = action_log_probs(behavior_policy_logits, actions)
    log_rhos = target_lp - behavior_lp
    vt = from_importance_weights(
        log_rhos=log_rhos.detach(),
        discounts=discounts,
        rewards=rewards,
        values=values.detach(),
        bootstrap_value=bootstrap_value.detach(),
        clip_rho_threshold=clip_rho_threshold,
        clip_pg_rho_threshold=clip_pg_rho_threshold,
    )
    return VTraceFromLogitsReturns(
        vs=vt.vs,
        pg_advantages=vt.pg_advantages,
        log_rhos=log_rhos,
        behavior_action_log_probs=behavior_lp,
        target_action_log_probs=target_lp,
    )
