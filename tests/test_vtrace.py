"""V-trace numerics: our implementation vs a direct evaluation of the
published formula (independent O(T^2) reference)."""
import torch

from moolib_amd.ops import vtrace


def naive_vtrace(log_rhos, discounts, rewards, values, bootstrap_value, rho_bar=1.0, c_bar=1.0):
    """Direct per-definition computation: v_s = V(x_s) + sum_{t>=s} gamma^{t-s}
    (prod_{i=s..t-1} c_i) delta_t ; O(T^2), used only as a test oracle."""
    T, B = rewards.shape
    rhos = log_rhos.exp()
    clipped_rhos = rhos.clamp(max=rho_bar)
    cs = rhos.clamp(max=c_bar)
    values_t1 = torch.cat([values[1:], bootstrap_value.unsqueeze(0)])
    deltas = clipped_rhos * (rewards + discounts * values_t1 - values)
    vs = torch.zeros_like(values)
    for s in range(T):
        total = torch.zeros(B)
        for t in range(s, T):
            prod = torch.ones(B)
            for i in range(s, t):
                prod = prod * discounts[i] * cs[i]
            total = total + prod * deltas[t]
        vs[s] = values[s] + total
    vs_t1 = torch.cat([vs[1:], bootstrap_value.unsqueeze(0)])
    pg_rhos = rhos.clamp(max=rho_bar)
    pg_adv = pg_rhos * (rewards + discounts * vs_t1 - values)
    return vs, pg_adv


class TestVtrace:
    def test_matches_naive(self):
        torch.manual_seed(0)
        T, B = 12, 5
        log_rhos = torch.randn(T, B) * 0.3
        discounts = (torch.rand(T, B) > 0.1).float() * 0.99
        rewards = torch.randn(T, B)
        values = torch.randn(T, B)
        bootstrap = torch.randn(B)
        got = vtrace.from_importance_weights(log_rhos, discounts, rewards, values, bootstrap)
        want_vs, want_pg = naive_vtrace(log_rhos, discounts, rewards, values, bootstrap)
        assert torch.allclose(got.vs, want_vs, atol=1e-5)
        assert torch.allclose(got.pg_advantages, want_pg, atol=1e-5)

    def test_on_policy_reduces_to_returns(self):
        """With rho=c=1 (on-policy), vs equals the discounted n-step return."""
        torch.manual_seed(1)
        T, B = 8, 3
        log_rhos = torch.zeros(T, B)
        discounts = torch.full((T, B), 0.9)
        rewards = torch.randn(T, B)
        values = torch.randn(T, B)
        bootstrap = torch.randn(B)
        got = vtrace.from_importance_weights(log_rhos, discounts, rewards, values, bootstrap)
        # on-policy: v_s = r_s + gamma v_{s+1}, v_T = bootstrap
        want = torch.empty(T, B)
        acc = bootstrap.clone()
        for t in reversed(range(T)):
            acc = rewards[t] + discounts[t] * acc
            want[t] = acc
        assert torch.allclose(got.vs, want, atol=1e-5)

    def test_from_logits_shapes(self):
        T, B, A = 6, 4, 9
        torch.manual_seed(2)
        out = vtrace.from_logits(
            behavior_policy_logits=torch.randn(T, B, A),
            target_policy_logits=torch.randn(T, B, A),
            actions=torch.randint(0, A, (T, B)),
            discounts=torch.full((T, B), 0.99),
            rewards=torch.randn(T, B),
            values=torch.randn(T, B),
            bootstrap_value=torch.randn(B),
        )
        assert out.vs.shape == (T, B)
        assert out.pg_advantages.shape == (T, B)
        assert out.log_rhos.shape == (T, B)
