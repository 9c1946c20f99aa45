"""Loss-function tests: hand-computed values and invariants."""
import math

import pytest
import torch

from stoix_amd.ops import losses as L


def test_ppo_clip_loss_values():
    # single sample: log-ratio 0 -> ratio 1 -> loss = -A
    lp = torch.tensor([0.0])
    blp = torch.tensor([0.0])
    adv = torch.tensor([2.0])
    assert abs(L.ppo_clip_loss(lp, blp, adv, 0.2).item() + 2.0) < 1e-6
    # big positive ratio with positive advantage clips at 1+eps
    lp = torch.tensor([1.0])
    out = L.ppo_clip_loss(lp, blp, adv, 0.2)
    assert abs(out.item() + 1.2 * 2.0) < 1e-6
    # big positive ratio with NEGATIVE advantage does NOT clip (pessimism)
    out = L.ppo_clip_loss(lp, blp, -adv, 0.2)
    assert abs(out.item() - math.exp(1.0) * 2.0) < 1e-4


def test_clipped_value_loss():
    pred = torch.tensor([2.0])
    behav = torch.tensor([0.0])
    tgt = torch.tensor([0.5])
    # clipped pred = 0 + clip(2, -0.5, 0.5) = 0.5; err=(2-0.5)^2=2.25
    # err_clipped=(0.5-0.5)^2=0 -> max = 2.25, 0.5*mean = 1.125
    assert abs(L.clipped_value_loss(pred, behav, tgt, 0.5).item() - 1.125) < 1e-6


def test_q_learning_loss():
    q_tm1 = torch.tensor([[1.0, 2.0]])
    a = torch.tensor([0])
    r = torch.tensor([1.0])
    d = torch.tensor([0.9])
    q_t = torch.tensor([[3.0, 5.0]])
    # target = 1 + .9*5 = 5.5; td = 5.5 - 1 = 4.5; mse/2 = 10.125
    assert abs(L.q_learning(q_tm1, a, r, d, q_t).item() - 0.5 * 4.5**2) < 1e-5


def test_double_q_learning_uses_selector_argmax():
    q_tm1 = torch.tensor([[1.0, 2.0]])
    a = torch.tensor([0])
    r = torch.tensor([0.0])
    d = torch.tensor([1.0])
    q_t_value = torch.tensor([[10.0, 20.0]])
    q_t_selector = torch.tensor([[5.0, 1.0]])  # argmax 0 -> value 10
    out = L.double_q_learning(q_tm1, q_t_value, a, r, d, q_t_selector)
    assert abs(out.item() - 0.5 * (10.0 - 1.0) ** 2) < 1e-5


def test_categorical_l2_project_identity_and_shift():
    z = torch.linspace(-1.0, 1.0, 5)
    probs = torch.tensor([[0.1, 0.2, 0.4, 0.2, 0.1]])
    # projecting onto itself is identity
    out = L.categorical_l2_project(z.unsqueeze(0), probs, z)
    torch.testing.assert_close(out, probs, rtol=1e-5, atol=1e-6)
    # shift by half a bin splits mass between neighbours
    out = L.categorical_l2_project(z.unsqueeze(0) + 0.25, probs, z)
    assert abs(out.sum().item() - 1.0) < 1e-5
    assert abs(out[0, 2].item() - (0.5 * 0.2 + 0.5 * 0.4)) < 1e-5
    # clamping at the edges keeps mass
    out = L.categorical_l2_project(z.unsqueeze(0) + 100.0, probs, z)
    assert abs(out[0, -1].item() - 1.0) < 1e-5


def test_categorical_double_q_learning_perfect_prediction():
    # if predicted dist equals the projected target, CE = entropy(target)
    atoms = torch.linspace(-1, 1, 11)
    B, A, N = 3, 2, 11
    g = torch.Generator().manual_seed(0)
    logits = torch.randn(B, A, N, generator=g)
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.zeros(B)
    d = torch.ones(B)
    sel = torch.randn(B, A, generator=g)
    loss = L.categorical_double_q_learning(logits, atoms, a, r, d, logits, atoms, sel)
    assert loss.item() > 0


def test_munchausen_reduces_to_soft_q():
    B, A = 4, 3
    g = torch.Generator().manual_seed(2)
    q = torch.randn(B, A, generator=g)
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g)
    d = torch.full((B,), 0.9)
    out = L.munchausen_q_learning(q, a, r, d, q, q, 0.03, 0.9, -1.0)
    assert torch.isfinite(out)


def test_quantile_regression_loss_zero_at_perfect():
    taus = torch.tensor([0.25, 0.75])
    src = torch.tensor([[1.0, 2.0]])
    tgt = torch.tensor([[1.0, 2.0]])
    out = L.quantile_regression_loss(src, taus, tgt, huber_param=1.0)
    # not exactly 0 (cross terms), but small and positive
    assert out.item() >= 0


def test_quantile_q_learning_runs():
    B, N, A = 4, 8, 3
    g = torch.Generator().manual_seed(4)
    dist = torch.randn(B, N, A, generator=g)
    taus = (torch.arange(N, dtype=torch.float32) + 0.5) / N
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g)
    d = torch.full((B,), 0.99)
    out = L.quantile_q_learning(dist, taus, a, r, d, dist, dist)
    assert torch.isfinite(out) and out.item() > 0


def test_dpo_loss_at_ratio_one():
    lp = torch.tensor([0.0])
    blp = torch.tensor([0.0])
    adv = torch.tensor([1.5])
    # ratio=1 -> drift terms 0 -> loss = -A
    out = L.dpo_loss(lp, blp, adv, alpha=2.0, beta=0.6)
    assert abs(out.item() + 1.5) < 1e-6


def test_ppo_penalty_kl_zero_at_same_policy():
    lp = torch.tensor([0.3, -0.2])
    loss, kl = L.ppo_penalty_loss(lp, lp, torch.tensor([1.0, 1.0]), 3.0)
    assert abs(kl.item()) < 1e-7
    assert abs(loss.item() + 1.0) < 1e-6


def test_categorical_l2_project_properties():
    """Property test for the Cramer projection (C51 backbone): on random
    source distributions and uniform target supports, the projection
    (a) preserves total mass, (b) preserves the MEAN whenever all source
    atoms lie inside [vmin, vmax] (the projection is linear-interpolating,
    so clamped-interior mass keeps its expectation), and (c) matches a
    naive per-atom two-bin split."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(0, 10_000))
    def run(seed):
        g = torch.Generator().manual_seed(seed)
        B, N, M = 4, 9, 11
        vmin, vmax = -5.0, 5.0
        z_q = torch.linspace(vmin, vmax, M)
        # interior source atoms (unsorted is fine) + random simplex weights
        z_p = (torch.rand(B, N, generator=g) * 9.0 - 4.5)
        w = torch.rand(B, N, generator=g) + 1e-3
        probs = w / w.sum(-1, keepdim=True)
        out = L.categorical_l2_project(z_p, probs, z_q)
        assert torch.all(out >= -1e-7)
        torch.testing.assert_close(
            out.sum(-1), torch.ones(B), rtol=1e-5, atol=1e-6
        )
        torch.testing.assert_close(
            (out * z_q).sum(-1), (probs * z_p).sum(-1), rtol=1e-4, atol=1e-5
        )
        # naive reference: split each atom's mass between its two bins
        dz = (vmax - vmin) / (M - 1)
        ref = torch.zeros(B, M)
        for bi in range(B):
            for n in range(N):
                pos = (float(z_p[bi, n]) - vmin) / dz
                lo, hi = int(pos // 1), min(int(pos // 1) + 1, M - 1)
                frac = pos - lo
                ref[bi, lo] += float(probs[bi, n]) * (1 - frac)
                ref[bi, hi] += float(probs[bi, n]) * frac
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)

    run()


def test_quantile_regression_loss_matches_scalar_reference():
    """Golden test: the vectorised pinball-Huber loss equals an explicit
    scalar triple loop over (batch, quantile, target-sample) pairs, for both
    huber_param > 0 and the pure-pinball huber_param = 0 branch."""
    g = torch.Generator().manual_seed(11)
    B, N, M = 3, 5, 7
    src = torch.randn(B, N, generator=g)
    tgt = torch.randn(B, M, generator=g)
    taus = torch.rand(N, generator=g).sort().values

    def scalar(huber):
        total = 0.0
        for b in range(B):
            per_q = 0.0
            for i in range(N):
                acc = 0.0
                for j in range(M):
                    delta = float(tgt[b, j]) - float(src[b, i])
                    w = abs(float(taus[i]) - (1.0 if delta < 0 else 0.0))
                    if huber > 0:
                        h = (
                            0.5 * delta * delta
                            if abs(delta) <= huber
                            else huber * (abs(delta) - 0.5 * huber)
                        )
                        acc += w * h / huber
                    else:
                        acc += w * abs(delta)
                per_q += acc / M
            total += per_q
        return total / B

    for huber in (1.0, 0.7, 0.0):
        out = L.quantile_regression_loss(src, taus, tgt, huber_param=huber)
        assert abs(out.item() - scalar(huber)) < 1e-5, huber


def test_munchausen_matches_scalar_reference():
    """Golden test: the Munchausen-DQN loss equals an explicit scalar
    computation (softmax policies via scipy logsumexp, clipped log-policy
    bonus, soft value target) on random instances."""
    from scipy.special import logsumexp

    g = torch.Generator().manual_seed(3)
    B, A = 5, 4
    q_tm1 = torch.randn(B, A, generator=g)
    q_target_tm1 = torch.randn(B, A, generator=g)
    q_t = torch.randn(B, A, generator=g)
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g)
    d = torch.full((B,), 0.9)
    tau, alpha, clip_min = 0.05, 0.9, -1.0

    total = 0.0
    for b in range(B):
        lt = (q_target_tm1[b] / tau).numpy()
        log_pi_tm1 = lt - logsumexp(lt)
        bonus = alpha * min(max(tau * log_pi_tm1[int(a[b])], clip_min), 0.0)
        ltt = (q_t[b] / tau).numpy()
        log_pi_t = ltt - logsumexp(ltt)
        import numpy as np

        pi_t = np.exp(log_pi_t)
        soft_v = float((pi_t * (q_t[b].numpy() - tau * log_pi_t)).sum())
        target = float(r[b]) + bonus + 0.9 * soft_v
        td = target - float(q_tm1[b, int(a[b])])
        total += 0.5 * td * td
    expect = total / B
    out = L.munchausen_q_learning(
        q_tm1, a, r, d, q_t, q_target_tm1, tau, alpha, clip_min
    )
    assert abs(out.item() - expect) < 1e-4


def test_categorical_double_q_learning_scalar_reference():
    """Golden test: the C51 double-Q loss equals a scalar reconstruction —
    pick argmax action from the selector, shift target atoms by r + d*z,
    two-bin project by hand, cross-entropy against the taken action's
    log-softmax."""
    import numpy as np
    from scipy.special import log_softmax, softmax

    g = torch.Generator().manual_seed(9)
    B, A, N = 4, 3, 7
    atoms = torch.linspace(-2.0, 2.0, N)
    logits_tm1 = torch.randn(B, A, N, generator=g)
    logits_t = torch.randn(B, A, N, generator=g)
    sel = torch.randn(B, A, generator=g)
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g) * 0.5
    d = torch.full((B,), 0.9)

    vmin, vmax = -2.0, 2.0
    dz = (vmax - vmin) / (N - 1)
    total = 0.0
    for b in range(B):
        best = int(sel[b].argmax())
        p_best = softmax(logits_t[b, best].numpy())
        proj = np.zeros(N)
        for n in range(N):
            z = float(r[b]) + 0.9 * float(atoms[n])
            z = min(max(z, vmin), vmax)
            pos = (z - vmin) / dz
            lo = int(pos // 1)
            hi = min(lo + 1, N - 1)
            frac = pos - lo
            proj[lo] += p_best[n] * (1 - frac)
            proj[hi] += p_best[n] * frac
        logp = log_softmax(logits_tm1[b, int(a[b])].numpy())
        total += -(proj * logp).sum()
    expect = total / B
    out = L.categorical_double_q_learning(
        logits_tm1, atoms, a, r, d, logits_t, atoms, sel
    )
    assert abs(out.item() - expect) < 1e-4


def test_categorical_td_learning_scalar_reference():
    """Golden test: distributional TD (D4PG critic) equals the scalar
    reconstruction — shift atoms by r + d*z, two-bin project, CE."""
    import numpy as np
    from scipy.special import log_softmax, softmax

    g = torch.Generator().manual_seed(6)
    B, N = 5, 9
    atoms = torch.linspace(-3.0, 3.0, N)
    logits_tm1 = torch.randn(B, N, generator=g)
    logits_t = torch.randn(B, N, generator=g)
    r = torch.randn(B, generator=g)
    d = torch.full((B,), 0.95)
    vmin, vmax = -3.0, 3.0
    dz = (vmax - vmin) / (N - 1)
    total = 0.0
    for b in range(B):
        p_t = softmax(logits_t[b].numpy())
        proj = np.zeros(N)
        for n in range(N):
            z = min(max(float(r[b]) + 0.95 * float(atoms[n]), vmin), vmax)
            pos = (z - vmin) / dz
            lo = int(pos // 1)
            hi = min(lo + 1, N - 1)
            frac = pos - lo
            proj[lo] += p_t[n] * (1 - frac)
            proj[hi] += p_t[n] * frac
        total += -(proj * log_softmax(logits_tm1[b].numpy())).sum()
    out = L.categorical_td_learning(logits_tm1, atoms, r, d, logits_t, atoms)
    assert abs(out.item() - total / B) < 1e-4


def test_dpo_and_penalty_losses_scalar_reference():
    """Golden tests at NON-trivial ratios: the DPO drift terms and the
    penalty loss's (r-1)-log r KL estimator match scalar math."""
    import math

    lp = torch.tensor([0.4, -0.3, 0.2])
    blp = torch.tensor([0.1, 0.1, 0.1])
    adv = torch.tensor([1.5, -2.0, 0.5])
    alpha, beta = 2.0, 0.6
    total = 0.0
    kl_total = 0.0
    for i in range(3):
        lr = float(lp[i] - blp[i])
        r = math.exp(lr)
        a = float(adv[i])
        if a >= 0:
            x = (r - 1.0) * a
            drift = max(0.0, x - alpha * math.tanh(x / alpha))
        else:
            x = lr * a
            drift = max(0.0, x - beta * math.tanh(x / beta))
        total += -(r * a - drift)
        kl_total += (r - 1.0) - lr
    out = L.dpo_loss(lp, blp, adv, alpha=alpha, beta=beta)
    assert abs(out.item() - total / 3) < 1e-5
    coef = 3.0
    loss, kl = L.ppo_penalty_loss(lp, blp, adv, coef)
    expect = 0.0
    for i in range(3):
        lr = float(lp[i] - blp[i])
        r = math.exp(lr)
        expect += -(r * float(adv[i]) - coef * ((r - 1.0) - lr))
    assert abs(loss.item() - expect / 3) < 1e-5
    assert abs(kl.item() - kl_total / 3) < 1e-6


def test_td_learning_and_quantile_q_learning_values():
    """Golden values: 1-step TD (plain + Huber branch) and QR-DQN double-Q
    selection — the loss must equal quantile_regression_loss applied to
    the taken action's quantiles vs the selector-argmax target dist."""
    v_tm1 = torch.tensor([1.0, -2.0])
    r = torch.tensor([0.5, 1.0])
    d = torch.tensor([0.9, 0.0])
    v_t = torch.tensor([2.0, 7.0])
    # plain: td = (0.5+1.8-1, 1+0-(-2)) = (1.3, 3.0); mse/2 mean
    expect = 0.5 * (1.3**2 + 3.0**2) / 2
    assert abs(L.td_learning(v_tm1, r, d, v_t).item() - expect) < 1e-6
    # huber delta=1: |1.3|>1 -> 1*(1.3-0.5); |3|>1 -> (3-0.5)
    expect_h = ((1.3 - 0.5) + (3.0 - 0.5)) / 2
    assert abs(L.td_learning(v_tm1, r, d, v_t, 1.0).item() - expect_h) < 1e-6

    g = torch.Generator().manual_seed(8)
    B, N, A = 4, 6, 3
    dist_tm1 = torch.randn(B, N, A, generator=g)
    dist_t = torch.randn(B, N, A, generator=g)
    sel = torch.randn(B, N, A, generator=g)
    taus = (torch.arange(N, dtype=torch.float32) + 0.5) / N
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g)
    d = torch.full((B,), 0.97)
    out = L.quantile_q_learning(dist_tm1, taus, a, r, d, sel, dist_t)
    best = sel.mean(dim=1).argmax(dim=-1)
    bidx = torch.arange(B)
    src = dist_tm1[bidx, :, a]
    tgt = r.unsqueeze(-1) + d.unsqueeze(-1) * dist_t[bidx, :, best]
    ref = L.quantile_regression_loss(src, taus, tgt, huber_param=1.0)
    assert abs(out.item() - ref.item()) < 1e-6
