"""CPU-fallback semantics of machin_amd.ops (the GPU twins are checked
against the same fp32 torch references in test_ops_gpu.py)."""
import pytest
import torch as t

from machin_amd import ops


class TestScans:
    def test_discounted_returns_1d(self):
        rew = t.tensor([1.0, 1.0, 1.0])
        term = t.tensor([0.0, 0.0, 1.0])
        out = ops.discounted_returns(rew, term, gamma=0.5)
        # R2=1 (terminal), R1=1+0.5*1, R0=1+0.5*1.5
        assert t.allclose(out, t.tensor([1.75, 1.5, 1.0]))

    def test_discounted_returns_bootstrap(self):
        rew = t.tensor([0.0, 0.0])
        term = t.tensor([0.0, 0.0])
        out = ops.discounted_returns(
            rew, term, gamma=0.9, bootstrap=t.tensor([10.0])
        )
        assert t.allclose(out, t.tensor([8.1, 9.0]))

    def test_gae_lambda_zero_is_td(self):
        T = 5
        rew = t.rand(T)
        val = t.rand(T)
        nxt = t.rand(T)
        term = t.zeros(T)
        out = ops.gae(rew, val, nxt, term, gamma=0.99, lam=0.0)
        expect = rew + 0.99 * nxt - val
        assert t.allclose(out, expect, atol=1e-6)

    def test_gae_lambda_one_is_mc_minus_v(self):
        T = 6
        rew = t.rand(T)
        term = t.zeros(T)
        term[-1] = 1.0
        val = t.rand(T)
        nxt = t.cat([val[1:], t.zeros(1)])
        adv = ops.gae(rew, val, nxt, term, gamma=0.9, lam=1.0)
        ret = ops.discounted_returns(rew, term, gamma=0.9)
        assert t.allclose(adv, ret - val, atol=1e-5)

    def test_nstep(self):
        rew = t.tensor([1.0, 2.0, 3.0, 4.0])
        term = t.tensor([0.0, 0.0, 0.0, 1.0])
        out = ops.nstep_returns(rew, term, gamma=0.5, n=2)
        assert t.allclose(out, t.tensor([2.0, 3.5, 5.0, 4.0]))

    def test_nstep_stops_at_terminal(self):
        rew = t.tensor([1.0, 5.0, 7.0])
        term = t.tensor([1.0, 0.0, 1.0])
        out = ops.nstep_returns(rew, term, gamma=0.9, n=3)
        assert t.allclose(out, t.tensor([1.0, 5.0 + 0.9 * 7.0, 7.0]))


class TestVtrace:
    def test_on_policy_reduces_to_td(self):
        """With rho=c=1 (same policy), vs should equal n-step TD(λ=1)
        bootstrapped value estimates."""
        T, B = 10, 4
        lp = t.randn(T, B) * 0.1
        rew = t.rand(T, B)
        val = t.rand(T, B)
        boot = t.rand(B)
        term = t.zeros(T, B)
        vs, pg = ops.vtrace(
            lp, lp, rew, val, boot, term, gamma=0.9,
            rho_clip=1.0, c_clip=1.0, pg_rho_clip=1.0,
        )
        # on-policy, rho=1: vs_t = r_t + gamma*vs_{t+1} (full MC to bootstrap)
        expect = t.empty(T, B)
        carry = boot.clone()
        for i in range(T - 1, -1, -1):
            carry = rew[i] + 0.9 * carry
            expect[i] = carry
        assert t.allclose(vs, expect, atol=1e-4)
        # pg advantage = r + gamma*vs_{t+1} - V
        vs_next = t.cat([vs[1:], boot.view(1, B)])
        assert t.allclose(pg, rew + 0.9 * vs_next - val, atol=1e-4)

    def test_terminal_masks_bootstrap(self):
        T, B = 3, 2
        lp = t.zeros(T, B)
        rew = t.ones(T, B)
        val = t.rand(T, B)
        boot = t.full((B,), 100.0)
        term = t.zeros(T, B)
        term[-1] = 1.0
        vs, _ = ops.vtrace(lp, lp, rew, val, boot, term, gamma=0.9)
        # terminal at last step: bootstrap never leaks in
        assert vs.max() < 5.0


class TestProjection:
    def test_projection_preserves_mass(self):
        B, A = 32, 51
        dist = t.softmax(t.randn(B, A), dim=1)
        rew = t.randn(B) * 5
        term = (t.rand(B) > 0.8).float()
        proj = ops.categorical_projection(dist, rew, term, 0.99, -10.0, 10.0)
        assert t.allclose(proj.sum(dim=1), t.ones(B), atol=1e-5)

    def test_projection_terminal_is_delta(self):
        """Terminal: all mass lands at the atom(s) nearest reward."""
        B, A = 1, 11  # support -5..5 step 1
        dist = t.full((B, A), 1.0 / A)
        rew = t.tensor([2.0])
        term = t.tensor([1.0])
        proj = ops.categorical_projection(dist, rew, term, 0.99, -5.0, 5.0)
        assert proj[0, 7].item() == pytest.approx(1.0)  # atom at +2

    def test_projection_interpolates(self):
        B, A = 1, 11
        dist = t.zeros(B, A)
        dist[0, 5] = 1.0  # atom 0
        rew = t.tensor([0.5])
        term = t.tensor([1.0])
        proj = ops.categorical_projection(dist, rew, term, 1.0, -5.0, 5.0)
        assert proj[0, 5].item() == pytest.approx(0.5)
        assert proj[0, 6].item() == pytest.approx(0.5)


class TestPolyak:
    def test_cpu_polyak(self):
        import torch.nn as nn

        a, b = nn.Linear(4, 4), nn.Linear(4, 4)
        from machin_amd.frame.algorithms.utils import soft_update

        a0 = [p.clone() for p in a.parameters()]
        soft_update(a, b, 0.25)
        for p, p0, ps in zip(a.parameters(), a0, b.parameters()):
            assert t.allclose(p, 0.75 * p0 + 0.25 * ps, atol=1e-6)
