"""RL loss functions (torch reference implementations).

Parity surface with /root/reference/stoix/utils/loss.py:
ppo_clip_loss :17-32, ppo_penalty_loss :35-47, dpo_loss :50-65,
clipped_value_loss :68-78, categorical_double_q_learning :81-103,
q_learning :106-124, double_q_learning :127-146, td_learning :149-163,
categorical_td_learning :166-187, munchausen_q_learning :190-223,
quantile_regression_loss / quantile_q_learning :226-314.

All are written against batched torch tensors; each is differentiable so the
update path can run under autograd (captured in a hip graph); the fused HIP
loss+grad kernels used by the Anakin fast path are numerics-tested against
these references (tests/test_losses.py).
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F

Tensor = torch.Tensor


# ---------------------------------------------------------------- policy-side


def ppo_clip_loss(
    pi_log_prob_t: Tensor,
    b_pi_log_prob_t: Tensor,
    gae_t: Tensor,
    epsilon: float,
) -> Tensor:
    """PPO clipped surrogate (Schulman et al. 2017)."""
    ratio = torch.exp(pi_log_prob_t - b_pi_log_prob_t)
    loss1 = ratio * gae_t
    loss2 = torch.clamp(ratio, 1.0 - epsilon, 1.0 + epsilon) * gae_t
    return -torch.minimum(loss1, loss2).mean()


def ppo_penalty_loss(
    pi_log_prob_t: Tensor,
    b_pi_log_prob_t: Tensor,
    gae_t: Tensor,
    kl_penalty_coef: float,
) -> Tuple[Tensor, Tensor]:
    """PPO with KL penalty instead of clipping: ratio*A - beta*KL(b||pi)
    (reference loss.py:35-47). Returns (loss, mean_approx_kl)."""
    log_ratio = pi_log_prob_t - b_pi_log_prob_t
    ratio = torch.exp(log_ratio)
    # Unbiased low-variance KL(b||pi) estimator: (r - 1) - log r
    approx_kl = (ratio - 1.0) - log_ratio
    loss = -(ratio * gae_t - kl_penalty_coef * approx_kl).mean()
    return loss, approx_kl.mean()


def dpo_loss(
    pi_log_prob_t: Tensor,
    b_pi_log_prob_t: Tensor,
    gae_t: Tensor,
    alpha: float,
    beta: float,
) -> Tensor:
    """Drift (DPO) policy loss (reference loss.py:50-65; arXiv:2310.19102).

    drift+ = relu((r-1)A - alpha*tanh((r-1)A/alpha)) for A>=0,
    drift- = relu(log(r)A - beta*tanh(log(r)A/beta)) for A<0;
    loss = -(r*A - drift).
    """
    log_ratio = pi_log_prob_t - b_pi_log_prob_t
    ratio = torch.exp(log_ratio)
    is_pos = (gae_t >= 0.0).to(gae_t.dtype)
    r1a = (ratio - 1.0) * gae_t
    drift_pos = F.relu(r1a - alpha * torch.tanh(r1a / alpha))
    lra = log_ratio * gae_t
    drift_neg = F.relu(lra - beta * torch.tanh(lra / beta))
    drift = is_pos * drift_pos + (1.0 - is_pos) * drift_neg
    return -(ratio * gae_t - drift).mean()


def clipped_value_loss(
    pred_value_t: Tensor,
    behavior_value_t: Tensor,
    targets_t: Tensor,
    epsilon: float,
) -> Tensor:
    """Clipped value loss (reference loss.py:68-78): max of clipped and
    unclipped squared errors, 0.5 * mean."""
    clipped = behavior_value_t + torch.clamp(
        pred_value_t - behavior_value_t, -epsilon, epsilon
    )
    err = (pred_value_t - targets_t) ** 2
    err_clipped = (clipped - targets_t) ** 2
    return 0.5 * torch.maximum(err, err_clipped).mean()


# ----------------------------------------------------------------- value-side


def _huber(x: Tensor, delta: float) -> Tensor:
    if delta <= 0:
        return 0.5 * x**2
    abs_x = x.abs()
    return torch.where(abs_x <= delta, 0.5 * x**2, delta * (abs_x - 0.5 * delta))


def q_learning(
    q_tm1: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    q_t: Tensor,
    huber_loss_parameter: float = 0.0,
) -> Tensor:
    """1-step Q-learning loss (reference loss.py:106-124).

    target = r + d * max_a q_t;  loss = huber/mse(target - q_tm1[a]).
    ``d_t`` = gamma * (1 - done).
    """
    q_a = q_tm1.gather(-1, a_tm1.long().unsqueeze(-1)).squeeze(-1)
    target = (r_t + d_t * q_t.max(dim=-1).values).detach()
    return _huber(target - q_a, huber_loss_parameter).mean()


def double_q_learning(
    q_tm1: Tensor,
    q_t_value: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    q_t_selector: Tensor,
    huber_loss_parameter: float = 0.0,
) -> Tensor:
    """Double Q-learning (reference loss.py:127-146): action argmax from the
    online net (selector), value from the target net."""
    q_a = q_tm1.gather(-1, a_tm1.long().unsqueeze(-1)).squeeze(-1)
    best_a = q_t_selector.argmax(dim=-1, keepdim=True)
    q_target_a = q_t_value.gather(-1, best_a).squeeze(-1)
    target = (r_t + d_t * q_target_a).detach()
    return _huber(target - q_a, huber_loss_parameter).mean()


def td_learning(
    v_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    v_t: Tensor,
    huber_loss_parameter: float = 0.0,
) -> Tensor:
    """1-step TD loss (reference loss.py:149-163)."""
    target = (r_t + d_t * v_t).detach()
    return _huber(target - v_tm1, huber_loss_parameter).mean()


def categorical_l2_project(
    z_p: Tensor,
    probs: Tensor,
    z_q: Tensor,
) -> Tensor:
    """Cramer projection of distribution (z_p, probs) onto support z_q
    (Bellemare et al. 2017; reference uses rlax.categorical_l2_project at
    loss.py:98,183). Batched: z_p [B, N], probs [B, N], z_q [M] -> [B, M]."""
    kq = z_q.shape[-1]
    vmin, vmax = z_q[0], z_q[-1]
    dz = (vmax - vmin) / (kq - 1)
    z_p = z_p.clamp(vmin, vmax)
    b = (z_p - vmin) / dz  # [B, N] fractional bin index
    lo = b.floor().clamp(0, kq - 1)
    hi = b.ceil().clamp(0, kq - 1)
    w_hi = b - lo
    w_lo = 1.0 - w_hi
    # when lo == hi (b integral), all weight to lo
    same = (lo == hi).to(probs.dtype)
    w_lo = w_lo + same * w_hi
    w_hi = w_hi * (1.0 - same)
    out = torch.zeros((*probs.shape[:-1], kq), dtype=probs.dtype, device=probs.device)
    out.scatter_add_(-1, lo.long(), probs * w_lo)
    out.scatter_add_(-1, hi.long(), probs * w_hi)
    return out


def categorical_double_q_learning(
    q_logits_tm1: Tensor,
    q_atoms_tm1: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    q_logits_t: Tensor,
    q_atoms_t: Tensor,
    q_t_selector: Tensor,
) -> Tensor:
    """C51 distributional double-Q loss (reference loss.py:81-103).

    Shift/scale target atoms by r + d*z, Cramer-project onto the fixed
    support, cross-entropy against the chosen action's logits.

    Shapes: q_logits [B, A, N], q_atoms [B, N] (or [N]), a/r/d [B],
    q_t_selector [B, A] mean values for argmax.
    """
    if q_atoms_t.dim() == 1:
        q_atoms_t = q_atoms_t.unsqueeze(0).expand(r_t.shape[0], -1)
    if q_atoms_tm1.dim() == 1:
        q_atoms_tm1 = q_atoms_tm1.unsqueeze(0).expand(r_t.shape[0], -1)
    target_z = r_t.unsqueeze(-1) + d_t.unsqueeze(-1) * q_atoms_t
    best_a = q_t_selector.argmax(dim=-1)
    probs_t = F.softmax(q_logits_t, dim=-1)
    p_best = probs_t.gather(1, best_a.view(-1, 1, 1).expand(-1, 1, probs_t.shape[-1])).squeeze(1)
    target = categorical_l2_project(target_z, p_best, q_atoms_tm1[0]).detach()
    logits_a = q_logits_tm1.gather(
        1, a_tm1.long().view(-1, 1, 1).expand(-1, 1, q_logits_tm1.shape[-1])
    ).squeeze(1)
    return -(target * F.log_softmax(logits_a, dim=-1)).sum(-1).mean()


def categorical_td_learning(
    v_logits_tm1: Tensor,
    v_atoms_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    v_logits_t: Tensor,
    v_atoms_t: Tensor,
) -> Tensor:
    """Distributional TD for categorical V/Q distributions
    (reference loss.py:166-187; D4PG critic)."""
    if v_atoms_t.dim() == 1:
        v_atoms_t = v_atoms_t.unsqueeze(0).expand(r_t.shape[0], -1)
    if v_atoms_tm1.dim() == 1:
        v_atoms_tm1 = v_atoms_tm1.unsqueeze(0).expand(r_t.shape[0], -1)
    target_z = r_t.unsqueeze(-1) + d_t.unsqueeze(-1) * v_atoms_t
    probs_t = F.softmax(v_logits_t, dim=-1)
    target = categorical_l2_project(target_z, probs_t, v_atoms_tm1[0]).detach()
    return -(target * F.log_softmax(v_logits_tm1, dim=-1)).sum(-1).mean()


def munchausen_q_learning(
    q_tm1: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    q_t: Tensor,
    q_target_tm1: Tensor,
    entropy_temperature: float,
    munchausen_coefficient: float,
    clip_value_min: float,
    huber_loss_parameter: float = 0.0,
) -> Tensor:
    """Munchausen DQN loss (Vieillard et al. 2020; reference loss.py:190-223).

    bonus = alpha * clip(tau * log_pi_target(a_tm1), min, 0)
    soft_target = sum_a pi_t(a) * (q_t(a) - tau * log pi_t(a))
    target = r + bonus + d * soft_target.
    """
    tau = entropy_temperature
    logits_target_tm1 = q_target_tm1 / tau
    log_pi_tm1 = F.log_softmax(logits_target_tm1, dim=-1)
    bonus = munchausen_coefficient * torch.clamp(
        tau * log_pi_tm1.gather(-1, a_tm1.long().unsqueeze(-1)).squeeze(-1),
        min=clip_value_min,
        max=0.0,
    )
    log_pi_t = F.log_softmax(q_t / tau, dim=-1)
    pi_t = log_pi_t.exp()
    soft_v = (pi_t * (q_t - tau * log_pi_t)).sum(-1)
    target = (r_t + bonus + d_t * soft_v).detach()
    q_a = q_tm1.gather(-1, a_tm1.long().unsqueeze(-1)).squeeze(-1)
    return _huber(target - q_a, huber_loss_parameter).mean()


def quantile_regression_loss(
    dist_src: Tensor,
    tau_src: Tensor,
    dist_target: Tensor,
    huber_param: float = 1.0,
) -> Tensor:
    """Quantile-regression (pinball) loss (reference loss.py:226-265).

    dist_src [B, N] quantile estimates with thresholds tau_src [N] (or [B, N]);
    dist_target [B, M] target samples.
    """
    if tau_src.dim() == 1:
        tau_src = tau_src.unsqueeze(0).expand(dist_src.shape[0], -1)
    # pairwise TD errors: target_j - src_i -> [B, N, M]
    delta = dist_target.detach().unsqueeze(1) - dist_src.unsqueeze(2)
    weight = (tau_src.unsqueeze(2) - (delta < 0).to(delta.dtype)).abs()
    if huber_param > 0:
        loss = _huber(delta, huber_param) / huber_param
    else:
        loss = delta.abs()
    return (weight * loss).mean(dim=2).sum(dim=1).mean()


def quantile_q_learning(
    dist_q_tm1: Tensor,
    tau_q_tm1: Tensor,
    a_tm1: Tensor,
    r_t: Tensor,
    d_t: Tensor,
    dist_q_t_selector: Tensor,
    dist_q_t: Tensor,
    huber_param: float = 1.0,
) -> Tensor:
    """QR-DQN loss (reference loss.py:268-314).

    dist_q [B, N, A]; selector distribution picks argmax over mean-quantile
    values; target dist = r + d * dist_q_t[:, :, a*].
    """
    q_a = dist_q_tm1.gather(
        2, a_tm1.long().view(-1, 1, 1).expand(-1, dist_q_tm1.shape[1], 1)
    ).squeeze(-1)
    best_a = dist_q_t_selector.mean(dim=1).argmax(dim=-1)
    target_dist = dist_q_t.gather(
        2, best_a.view(-1, 1, 1).expand(-1, dist_q_t.shape[1], 1)
    ).squeeze(-1)
    target = (r_t.unsqueeze(-1) + d_t.unsqueeze(-1) * target_dist).detach()
    return quantile_regression_loss(q_a, tau_q_tm1, target, huber_param)
