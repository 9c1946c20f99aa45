"""Multistep return / advantage estimators (torch reference implementations).

Functional parity surface with /root/reference/stoix/utils/multistep.py
(batch_truncated_generalized_advantage_estimation :14-145,
batch_n_step_bootstrapped_returns :148-207, retrace :210-311,
batch_lambda_returns / batch_discounted_returns :314-449,
importance_corrected_td_errors :452-530, batch_q_lambda :533-569) and the
vtrace estimator the reference takes from rlax
(systems/impala/sebulba/ff_impala.py:426-440).

All functions are **time-major**: inputs are ``[T, B]`` (or ``[T, B, ...]``).
Reverse time scans are implemented as explicit Python loops over T — correct
and fast enough on CPU for tests; the GPU fast path is the HIP reverse-scan
kernel in ``stoix_amd/ops/hip/scan.hip`` (dispatched in ``stoix_amd.ops``),
which parallelises over B and runs the T recursion in-kernel.

Conventions:
  * ``discount_t`` already includes gamma: ``discount_t = gamma * (1 - done)``
    unless a function takes ``gamma`` explicitly (its docstring says so).
  * termination => discount 0; truncation => discount gamma with a separate
    ``truncation_t`` mask that resets accumulators without zeroing bootstrap.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

Tensor = torch.Tensor


def batch_truncated_generalized_advantage_estimation(
    r_t: Tensor,
    discount_t: Tensor,
    lambda_: float,
    v_tm1: Tensor,
    v_t: Tensor,
    truncation_t: Optional[Tensor] = None,
    standardize_advantages: bool = False,
) -> Tuple[Tensor, Tensor]:
    """Truncation-aware GAE (reference multistep.py:14-145).

    adv_t = delta_t + discount_t * lambda * (1 - trunc_t) * adv_{t+1}
    delta_t = r_t + discount_t * v_t - v_tm1

    Args:
        r_t: rewards [T, B].
        discount_t: gamma * (1 - done) [T, B] (0 exactly at termination).
        lambda_: GAE lambda.
        v_tm1: V(s_t) [T, B].
        v_t: V(s_{t+1}) [T, B] — the *bootstrap* value of the true next
            observation (extras["next_obs"] under autoreset).
        truncation_t: bool [T, B]; resets the accumulator across truncation
            boundaries (the delta still bootstraps via v_t).
    Returns:
        (advantages [T, B], target_values = advantages + v_tm1).
    """
    T = r_t.shape[0]
    if r_t.is_cuda:
        from stoix_amd import ops as _ops

        e = _ops.ext(required=True)
        adv = torch.empty_like(r_t)
        targets = torch.empty_like(r_t)
        tr = (
            truncation_t.to(torch.uint8).contiguous()
            if truncation_t is not None
            else torch.empty(0, dtype=torch.uint8, device=r_t.device)
        )
        e.gae(
            r_t.contiguous(),
            discount_t.contiguous(),
            v_tm1.contiguous(),
            v_t.contiguous(),
            tr,
            adv,
            targets,
            float(lambda_),
        )
        if standardize_advantages:
            adv = (adv - adv.mean()) / (adv.std(unbiased=False) + 1e-8)
        return adv, targets
    if truncation_t is None:
        cont = torch.ones_like(discount_t)
    else:
        cont = 1.0 - truncation_t.to(r_t.dtype)
    delta = r_t + discount_t * v_t - v_tm1
    adv = torch.empty_like(delta)
    acc = torch.zeros_like(delta[0])
    for t in range(T - 1, -1, -1):
        acc = delta[t] + discount_t[t] * lambda_ * cont[t] * acc
        adv[t] = acc
    targets = adv + v_tm1
    if standardize_advantages:
        adv = (adv - adv.mean()) / (adv.std(unbiased=False) + 1e-8)
    return adv, targets


def batch_n_step_bootstrapped_returns(
    r_t: Tensor,
    discount_t: Tensor,
    v_t: Tensor,
    n: int,
    lambda_t: float = 1.0,
) -> Tensor:
    """n-step bootstrapped returns (reference multistep.py:148-207).

    For each t: G_t = r_t + d_t*(r_{t+1} + d_{t+1}*(... + d_{t+n-1} *
    [(1-λ) v_{t+n-1-mix} ... standard mixed bootstrap] )) with the window
    clipped at the sequence end (where it bootstraps from v_T-1).

    Args:
        r_t, discount_t, v_t: [T, B]; v_t[t] = V(s_{t+1}).
    Returns:
        targets [T, B].
    """
    T = r_t.shape[0]
    # Pad: treat steps beyond the end by repeating the final bootstrap value
    # with zero extra reward (standard rlax behaviour).
    pad = n - 1
    r = torch.cat([r_t, torch.zeros((pad, *r_t.shape[1:]), dtype=r_t.dtype, device=r_t.device)], 0)
    d = torch.cat([discount_t, torch.ones((pad, *r_t.shape[1:]), dtype=r_t.dtype, device=r_t.device)], 0)
    v = torch.cat([v_t, v_t[-1:].expand(pad, *v_t.shape[1:])], 0)
    targets = v[n - 1 : n - 1 + T].clone()
    for i in range(n - 1, -1, -1):
        r_i = r[i : i + T]
        d_i = d[i : i + T]
        v_i = v[i : i + T]
        targets = r_i + d_i * ((1 - lambda_t) * v_i + lambda_t * targets)
    return targets


def batch_lambda_returns(
    r_t: Tensor,
    discount_t: Tensor,
    v_t: Tensor,
    lambda_: float = 1.0,
) -> Tensor:
    """TD(lambda) returns (reference multistep.py:314-390).

    G_t = r_t + discount_t * ((1-λ) v_t + λ G_{t+1}), G_T = r_T + d_T v_T.
    v_t[t] = V(s_{t+1}).
    """
    T = r_t.shape[0]
    if r_t.is_cuda:
        from stoix_amd import ops as _ops

        e = _ops.ext(required=True)
        out = torch.empty_like(r_t)
        e.lambda_returns(r_t.contiguous(), discount_t.contiguous(), v_t.contiguous(), out, float(lambda_))
        return out
    out = torch.empty_like(r_t)
    acc = v_t[-1]
    for t in range(T - 1, -1, -1):
        acc = r_t[t] + discount_t[t] * ((1 - lambda_) * v_t[t] + lambda_ * acc)
        out[t] = acc
    return out


def batch_discounted_returns(
    r_t: Tensor,
    discount_t: Tensor,
    v_t: Tensor,
) -> Tensor:
    """Monte-Carlo discounted returns bootstrapped at the final step
    (reference multistep.py:393-449). Equivalent to lambda returns at λ=1."""
    return batch_lambda_returns(r_t, discount_t, v_t, lambda_=1.0)


def batch_general_off_policy_returns_from_q_and_v(
    q_t: Tensor,
    v_t: Tensor,
    r_t: Tensor,
    discount_t: Tensor,
    c_t: Tensor,
) -> Tensor:
    """General off-policy corrected returns (reference multistep.py:210-268).

    G_t = r_t + d_t * (v_t - c_t * q_t + c_t * G_{t+1}) with the recursion
    seeded by G_{T} = v_{T} (i.e. the last target bootstraps from v only).

    Shapes: q_t, v_t [T, B] where q_t[t]=Q(s_{t+1},a_{t+1}), v_t[t]=V(s_{t+1});
    c_t [T-?, B] trace coefficients aligned with q_t; here all are [T, B] and
    c_t[-1] is unused.
    """
    T = r_t.shape[0]
    if r_t.is_cuda:
        from stoix_amd import ops as _ops

        e = _ops.ext(required=True)
        out = torch.empty_like(r_t)
        e.offpolicy_returns(
            q_t.contiguous(), v_t.contiguous(), r_t.contiguous(),
            discount_t.contiguous(), c_t.contiguous(), out,
        )
        return out
    out = torch.empty_like(r_t)
    g = r_t[T - 1] + discount_t[T - 1] * v_t[T - 1]
    out[T - 1] = g
    for t in range(T - 2, -1, -1):
        g = r_t[t] + discount_t[t] * (v_t[t] - c_t[t] * q_t[t] + c_t[t] * g)
        out[t] = g
    return out


def batch_retrace_continuous(
    q_tm1: Tensor,
    q_t: Tensor,
    v_t: Tensor,
    r_t: Tensor,
    discount_t: Tensor,
    log_rhos: Tensor,
    lambda_: float = 1.0,
) -> Tensor:
    """Retrace(λ) targets for continuous control (reference multistep.py:270-311).

    c_t = λ * min(1, ρ_t); targets from
    ``batch_general_off_policy_returns_from_q_and_v``; returns the regression
    target for q_tm1 (i.e. the corrected return, not the TD error).
    """
    c_t = lambda_ * torch.clamp(torch.exp(log_rhos), max=1.0)
    return batch_general_off_policy_returns_from_q_and_v(q_t, v_t, r_t, discount_t, c_t)


def importance_corrected_td_errors(
    r_t: Tensor,
    discount_t: Tensor,
    rho_tm1: Tensor,
    lambda_: float,
    values: Tensor,
) -> Tensor:
    """Per-decision importance-weighted multistep TD errors
    (reference multistep.py:452-530; used by V-MPO).

    e_t = ρ̄_t * (δ_t + discount_t * λ * e_{t+1}),  ρ̄_t = min(1, ρ_t),
    δ_t = r_t + discount_t * values[t+1] - values[t].

    Args:
        values: [T+1, B] state values (the extra row is the bootstrap).
    Returns:
        errors [T, B].
    """
    T = r_t.shape[0]
    rho = torch.clamp(rho_tm1, max=1.0)
    v_tm1, v_t = values[:-1], values[1:]
    delta = r_t + discount_t * v_t - v_tm1
    out = torch.empty_like(delta)
    acc = torch.zeros_like(delta[0])
    for t in range(T - 1, -1, -1):
        acc = rho[t] * (delta[t] + discount_t[t] * lambda_ * acc)
        out[t] = acc
    return out


def batch_q_lambda(
    q_tm1: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    discount_t: Tensor,
    q_t: Tensor,
    lambda_: float,
) -> Tensor:
    """Peng's Q(λ) targets (reference multistep.py:533-569; PQN).

    G_t = r_t + d_t * ((1-λ) max_a q_t + λ G_{t+1}), G seeded at max_a q_T.

    Args:
        q_tm1: [T, B, A] Q(s_t, ·) (only used for dtype/shape parity).
        a_tm1: [T, B] actions (unused in target computation, kept for parity).
        q_t: [T, B, A] Q(s_{t+1}, ·).
    Returns:
        targets [T, B].
    """
    v_t = q_t.max(dim=-1).values
    return batch_lambda_returns(r_t, discount_t, v_t, lambda_)


def vtrace_td_error_and_advantage(
    v_tm1: Tensor,
    v_t: Tensor,
    r_t: Tensor,
    discount_t: Tensor,
    rho_tm1: Tensor,
    lambda_: float = 1.0,
    clip_rho_threshold: float = 1.0,
    clip_pg_rho_threshold: float = 1.0,
) -> Tuple[Tensor, Tensor, Tensor]:
    """V-trace (IMPALA, Espeholt et al. 2018) — parity with the rlax call at
    reference systems/impala/sebulba/ff_impala.py:426-440.

    Returns (errors = vs - v_tm1, pg_advantage, q_estimate) each [T, B].
    """
    if r_t.is_cuda:
        from stoix_amd import ops as _ops

        e = _ops.ext(required=True)
        errors = torch.empty_like(r_t)
        pg_adv = torch.empty_like(r_t)
        e.vtrace(
            v_tm1.contiguous(), v_t.contiguous(), r_t.contiguous(),
            discount_t.contiguous(), rho_tm1.contiguous(), errors, pg_adv,
            float(lambda_), float(clip_rho_threshold), float(clip_pg_rho_threshold),
        )
        vs_t = torch.cat([errors[1:] + v_tm1[1:], v_t[-1:]], dim=0)
        q_estimate = r_t + discount_t * vs_t
        return errors, pg_adv, q_estimate
    rho_clip = torch.clamp(rho_tm1, max=clip_rho_threshold)
    c_t = lambda_ * torch.clamp(rho_tm1, max=1.0)
    delta = rho_clip * (r_t + discount_t * v_t - v_tm1)
    T = r_t.shape[0]
    err = torch.empty_like(delta)
    acc = torch.zeros_like(delta[0])
    for t in range(T - 1, -1, -1):
        acc = delta[t] + discount_t[t] * c_t[t] * acc
        err[t] = acc
    vs = err + v_tm1
    vs_t = torch.cat([vs[1:], v_t[-1:]], dim=0)
    pg_rho = torch.clamp(rho_tm1, max=clip_pg_rho_threshold)
    q_estimate = r_t + discount_t * vs_t
    pg_advantage = pg_rho * (q_estimate - v_tm1)
    return vs - v_tm1, pg_advantage, q_estimate
