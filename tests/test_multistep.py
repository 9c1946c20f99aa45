"""Golden-value tests for multistep estimators.

Mirrors the reference's test strategy (/root/reference/stoix/tests/
multistep_test.py: hand-computed truncated-GAE expectations for
lambda in {0, 0.5, 0.9, 1}, termination vs truncation vs both).
Expectations here are computed by independent scalar recurrences coded
inline (not by the functions under test).
"""
import math

import pytest
import torch

from stoix_amd.ops import multistep as ms


def scalar_gae(rewards, discounts, lam, v_tm1, v_t, trunc):
    T = len(rewards)
    adv = [0.0] * T
    acc = 0.0
    for t in range(T - 1, -1, -1):
        delta = rewards[t] + discounts[t] * v_t[t] - v_tm1[t]
        acc = delta + discounts[t] * lam * (0.0 if trunc[t] else 1.0) * acc
        adv[t] = acc
    return adv


def _to_tb(x):
    return torch.tensor(x, dtype=torch.float32).unsqueeze(1)


@pytest.mark.parametrize("lam", [0.0, 0.5, 0.9, 1.0])
def test_gae_matches_scalar_recurrence(lam):
    g = torch.Generator().manual_seed(0)
    T = 12
    r = torch.randn(T, generator=g).tolist()
    v_tm1 = torch.randn(T, generator=g).tolist()
    v_t = torch.randn(T, generator=g).tolist()
    done = [False, False, True, False, False, False, False, True, False, False, False, False]
    trunc = [False, False, False, False, True, False, False, False, False, False, True, False]
    gamma = 0.97
    disc = [0.0 if d else gamma for d in done]
    expected = scalar_gae(r, disc, lam, v_tm1, v_t, trunc)
    adv, targets = ms.batch_truncated_generalized_advantage_estimation(
        _to_tb(r), _to_tb(disc), lam, _to_tb(v_tm1), _to_tb(v_t), _to_tb([float(t) for t in trunc]).bool()
    )
    torch.testing.assert_close(adv.squeeze(1), torch.tensor(expected), rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(targets, adv + _to_tb(v_tm1))


def test_gae_termination_blocks_bootstrap_and_accumulation():
    # single termination mid-sequence: advantage before it must not see
    # anything after it (discount 0 kills both delta bootstrap and recursion)
    r = [1.0, 1.0, 1.0, 1.0]
    v_tm1 = [0.0, 0.0, 0.0, 0.0]
    v_t = [5.0, 5.0, 5.0, 5.0]
    done_disc = [0.9, 0.0, 0.9, 0.9]
    adv, _ = ms.batch_truncated_generalized_advantage_estimation(
        _to_tb(r), _to_tb(done_disc), 1.0, _to_tb(v_tm1), _to_tb(v_t)
    )
    # t=1: delta = 1 + 0*5 - 0 = 1; no accumulation from t>=2
    assert abs(adv[1, 0].item() - 1.0) < 1e-6
    # t=0 sees only t=1: delta0 + 0.9*1*adv1 = (1+0.9*5) + 0.9 = 6.4
    assert abs(adv[0, 0].item() - 6.4) < 1e-6


def test_gae_truncation_keeps_bootstrap_resets_accumulator():
    r = [1.0, 1.0, 1.0]
    v_tm1 = [0.0, 0.0, 0.0]
    v_t = [5.0, 5.0, 5.0]
    disc = [0.9, 0.9, 0.9]
    trunc = [False, True, False]
    adv, _ = ms.batch_truncated_generalized_advantage_estimation(
        _to_tb(r), _to_tb(disc), 1.0, _to_tb(v_tm1), _to_tb(v_t), _to_tb([0.0, 1.0, 0.0]).bool()
    )
    # t=1 truncated: delta = 1 + 0.9*5 = 5.5 (bootstrap kept), acc resets
    assert abs(adv[1, 0].item() - 5.5) < 1e-6
    # t=0: delta0 + 0.9*adv1 = 5.5 + 0.9*5.5 = 10.45
    assert abs(adv[0, 0].item() - 10.45) < 1e-6


def test_gae_standardize():
    g = torch.Generator().manual_seed(1)
    r = torch.randn(8, 4, generator=g)
    d = torch.full((8, 4), 0.99)
    v = torch.randn(8, 4, generator=g)
    vb = torch.randn(8, 4, generator=g)
    adv, _ = ms.batch_truncated_generalized_advantage_estimation(
        r, d, 0.95, v, vb, standardize_advantages=True
    )
    assert abs(adv.mean().item()) < 1e-5
    assert abs(adv.std(unbiased=False).item() - 1.0) < 1e-4


def test_lambda_returns_known_values():
    # lambda=1 == discounted MC bootstrapped from final value
    r = [1.0, 2.0, 3.0]
    d = [0.5, 0.5, 0.5]
    v = [10.0, 10.0, 4.0]
    out = ms.batch_lambda_returns(_to_tb(r), _to_tb(d), _to_tb(v), 1.0)
    # G2 = 3 + .5*4 = 5; G1 = 2 + .5*5 = 4.5; G0 = 1 + .5*4.5 = 3.25
    torch.testing.assert_close(out.squeeze(1), torch.tensor([3.25, 4.5, 5.0]))
    # lambda=0 == one-step TD targets
    out0 = ms.batch_lambda_returns(_to_tb(r), _to_tb(d), _to_tb(v), 0.0)
    torch.testing.assert_close(out0.squeeze(1), torch.tensor([6.0, 7.0, 5.0]))


def test_n_step_returns():
    r = [1.0, 1.0, 1.0, 1.0]
    d = [0.9, 0.9, 0.9, 0.9]
    v = [2.0, 2.0, 2.0, 2.0]
    out = ms.batch_n_step_bootstrapped_returns(_to_tb(r), _to_tb(d), _to_tb(v), n=2)
    # G0 = r0 + d0*(r1 + d1*v1) = 1 + .9*(1 + .9*2) = 3.52
    assert abs(out[0, 0].item() - 3.52) < 1e-6
    # G2 = r2 + d2*(r3 + d3*v3) = same window
    assert abs(out[2, 0].item() - 3.52) < 1e-6
    # G3 clipped at end: r3 + d3*v3 = 2.8
    assert abs(out[3, 0].item() - 2.8) < 1e-6


def test_n_step_termination_zeroes_tail():
    r = [1.0, 1.0, 1.0]
    d = [0.9, 0.0, 0.9]
    v = [2.0, 2.0, 2.0]
    out = ms.batch_n_step_bootstrapped_returns(_to_tb(r), _to_tb(d), _to_tb(v), n=3)
    # G0 = r0 + d0*(r1 + 0*(...)) = 1 + .9*1 = 1.9
    assert abs(out[0, 0].item() - 1.9) < 1e-6


def test_retrace_matches_scalar():
    g = torch.Generator().manual_seed(3)
    T = 6
    q = torch.randn(T, 1, generator=g)
    v = torch.randn(T, 1, generator=g)
    r = torch.randn(T, 1, generator=g)
    d = torch.full((T, 1), 0.95)
    log_rho = torch.randn(T, 1, generator=g) * 0.5
    lam = 0.9
    c = lam * torch.clamp(log_rho.exp(), max=1.0)
    out = ms.batch_retrace_continuous(q, q, v, r, d, log_rho, lam)
    # scalar recurrence
    gacc = (r[T - 1] + d[T - 1] * v[T - 1]).item()
    exp = [0.0] * T
    exp[T - 1] = gacc
    for t in range(T - 2, -1, -1):
        gacc = (r[t] + d[t] * (v[t] - c[t] * q[t] + c[t] * gacc)).item()
        exp[t] = gacc
    torch.testing.assert_close(out.squeeze(1), torch.tensor(exp), rtol=1e-5, atol=1e-5)


def test_q_lambda():
    qs = torch.tensor([[[1.0, 3.0]], [[2.0, 0.0]], [[1.0, 5.0]]])  # [T,1,A]
    r = _to_tb([1.0, 1.0, 1.0])
    d = _to_tb([0.5, 0.5, 0.5])
    a = torch.zeros(3, 1, dtype=torch.long)
    out = ms.batch_q_lambda(qs, a, r, d, qs, lambda_=1.0)
    # max_a q = [3, 2, 5]; G2 = 1+.5*5=3.5; G1 = 1+.5*3.5=2.75; G0 = 1+.5*2.75=2.375
    torch.testing.assert_close(out.squeeze(1), torch.tensor([2.375, 2.75, 3.5]))


def test_vtrace_on_policy_equals_gae_lambda1():
    """With rho=1 V-trace errors reduce to lambda-1 GAE on v."""
    g = torch.Generator().manual_seed(5)
    T = 8
    v_tm1 = torch.randn(T, 2, generator=g)
    v_t = torch.cat([v_tm1[1:], torch.randn(1, 2, generator=g)], 0)
    r = torch.randn(T, 2, generator=g)
    d = torch.full((T, 2), 0.9)
    rho = torch.ones(T, 2)
    errors, pg_adv, _ = ms.vtrace_td_error_and_advantage(v_tm1, v_t, r, d, rho, lambda_=1.0)
    adv, _ = ms.batch_truncated_generalized_advantage_estimation(r, d, 1.0, v_tm1, v_t)
    torch.testing.assert_close(errors, adv, rtol=1e-4, atol=1e-5)


def test_importance_corrected_td_errors_on_policy():
    g = torch.Generator().manual_seed(6)
    T = 5
    values = torch.randn(T + 1, 3, generator=g)
    r = torch.randn(T, 3, generator=g)
    d = torch.full((T, 3), 0.8)
    rho = torch.ones(T, 3)
    out = ms.importance_corrected_td_errors(r, d, rho, 0.9, values)
    adv, _ = ms.batch_truncated_generalized_advantage_estimation(
        r, d, 0.9, values[:-1], values[1:]
    )
    torch.testing.assert_close(out, adv, rtol=1e-4, atol=1e-5)


def test_gae_matches_bruteforce_property():
    """Property test: the GAE recursion (with truncation resets) equals a
    brute-force O(T^2) evaluation of the lambda-advantage sum on random
    problem instances."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from stoix_amd.ops import multistep as ms

    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(0, 10_000))
    def run(seed):
        g = torch.Generator().manual_seed(seed)
        T, B = int(torch.randint(2, 12, (1,), generator=g)), 3
        r = torch.randn(T, B, generator=g)
        d = (torch.rand(T, B, generator=g) > 0.2).float() * 0.93
        v = torch.randn(T, B, generator=g)
        vb = torch.randn(T, B, generator=g)
        trunc = torch.rand(T, B, generator=g) > 0.8
        lam = float(torch.rand(1, generator=g))
        adv, tgt = ms.batch_truncated_generalized_advantage_estimation(
            r, d, lam, v, vb, truncation_t=trunc
        )
        # brute force: A_t = sum_{k>=t} (prod coef) * delta_k with the
        # accumulator killed after a truncated step
        delta = r + d * vb - v
        for b in range(B):
            for t in range(T):
                acc, coef = 0.0, 1.0
                for k in range(t, T):
                    acc += coef * float(delta[k, b])
                    coef *= float(d[k, b]) * lam * (0.0 if bool(trunc[k, b]) else 1.0)
                    if coef == 0.0:
                        break
                assert abs(acc - float(adv[t, b])) < 1e-4, (seed, t, b)
                assert abs(acc + float(v[t, b]) - float(tgt[t, b])) < 1e-4

    run()


def test_vtrace_matches_bruteforce_property():
    """Property test: the v-trace recurrence equals the paper's closed form
    (Espeholt et al. 2018, eq. 1) evaluated brute-force on random instances:

      vs_s = V_s + sum_{t>=s} (prod_{i=s}^{t-1} gamma_i c_i) * delta_t,
      delta_t = clip(rho_t) (r_t + gamma_t V_{t+1} - V_t),
      c_i = lambda * clip(rho_i, 1),

    and pg_adv_t = clip(rho_t, pg_thresh) (r_t + gamma_t vs_{t+1} - V_t).
    """
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from stoix_amd.ops import multistep as ms

    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(0, 10_000))
    def run(seed):
        g = torch.Generator().manual_seed(seed)
        T, B = int(torch.randint(2, 12, (1,), generator=g)), 3
        r = torch.randn(T, B, generator=g)
        d = (torch.rand(T, B, generator=g) > 0.2).float() * 0.95
        v_tm1 = torch.randn(T, B, generator=g)
        v_t = torch.randn(T, B, generator=g)
        # off-policy: log-ratio noise gives rhos straddling the clip at 1
        rho = torch.exp(0.7 * torch.randn(T, B, generator=g))
        lam = float(torch.rand(1, generator=g))
        errors, pg_adv, q_est = ms.vtrace_td_error_and_advantage(
            v_tm1, v_t, r, d, rho, lambda_=lam
        )
        delta = rho.clamp(max=1.0) * (r + d * v_t - v_tm1)
        c = lam * rho.clamp(max=1.0)
        vs = torch.empty(T, B)
        for b in range(B):
            for s in range(T):
                acc, coef = 0.0, 1.0
                for t in range(s, T):
                    acc += coef * float(delta[t, b])
                    coef *= float(d[t, b]) * float(c[t, b])
                vs[s, b] = float(v_tm1[s, b]) + acc
        torch.testing.assert_close(errors, vs - v_tm1, rtol=1e-4, atol=1e-5)
        vs_t = torch.cat([vs[1:], v_t[-1:]], dim=0)
        expect_q = r + d * vs_t
        torch.testing.assert_close(q_est, expect_q, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(
            pg_adv, rho.clamp(max=1.0) * (expect_q - v_tm1), rtol=1e-4, atol=1e-5
        )

    run()


def test_retrace_matches_bruteforce_property():
    """Property test: the general off-policy return recursion (retrace's
    core) equals its brute-force expansion

      G_s = sum_{t=s}^{T-1} (prod_{i=s}^{t-1} d_i c_i) * b_t,
      b_t = r_t + d_t (v_t - c_t q_t)  for t < T-1,
      b_{T-1} = r_{T-1} + d_{T-1} v_{T-1}   (c at the horizon unused),

    on random off-policy instances."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from stoix_amd.ops import multistep as ms

    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(0, 10_000))
    def run(seed):
        g = torch.Generator().manual_seed(seed)
        T, B = int(torch.randint(2, 12, (1,), generator=g)), 3
        r = torch.randn(T, B, generator=g)
        d = (torch.rand(T, B, generator=g) > 0.2).float() * 0.95
        q = torch.randn(T, B, generator=g)
        v = torch.randn(T, B, generator=g)
        log_rhos = 0.7 * torch.randn(T, B, generator=g)
        lam = float(torch.rand(1, generator=g))
        out = ms.batch_retrace_continuous(
            torch.randn(T, B, generator=g), q, v, r, d, log_rhos, lambda_=lam
        )
        c = lam * torch.exp(log_rhos).clamp(max=1.0)
        for b in range(B):
            for s in range(T):
                acc, coef = 0.0, 1.0
                for t in range(s, T):
                    if t == T - 1:
                        bt = float(r[t, b]) + float(d[t, b]) * float(v[t, b])
                    else:
                        bt = float(r[t, b]) + float(d[t, b]) * (
                            float(v[t, b]) - float(c[t, b]) * float(q[t, b])
                        )
                    acc += coef * bt
                    coef *= float(d[t, b]) * float(c[t, b])
                assert abs(acc - float(out[s, b])) < 1e-4, (seed, s, b)

    run()


def test_discounted_returns_scalar():
    """Golden values for the plain discounted-return scan (REINFORCE)."""
    r = _to_tb([1.0, 2.0, 3.0])
    d = _to_tb([0.5, 0.5, 0.0])
    v = _to_tb([0.0, 0.0, 0.0])
    out = ms.batch_discounted_returns(r, d, v)
    # G2 = 3; G1 = 2 + 0.5*3 = 3.5; G0 = 1 + 0.5*3.5 = 2.75
    torch.testing.assert_close(out, _to_tb([2.75, 3.5, 3.0]))
