"""Validates the closed-form backward used by ppo_head_loss_kernel
(stoix_amd/ops/csrc/mlp.hip) against torch autograd on the eager formulas.

The kernel computes, per row: head dots (loc/scale_pre/value), tanh-normal
log-prob of the stored action, PPO clip loss, clipped value loss, MC
entropy, and the analytic gradients d(total)/d{loc, scale_pre, v}. This test
re-implements exactly that math in fp64 torch (`fused_head_reference`) and
checks both the forward values and the gradients against autograd through
the same eager ops the non-fused path uses (losses.py + distributions.py).
"""
from __future__ import annotations

import math

import pytest
import torch
import torch.nn.functional as F

torch.manual_seed(0)


def fused_head_reference(loc, spre, v_pred, action, old_logp, old_value, adv,
                         targets, eps_ent, clip_eps, ent_coef, vf_coef,
                         min_scale, aff_scale, aff_shift):
    """The kernel's forward math (differentiable; autograd supplies the
    reference gradients the kernel's closed forms must match)."""
    log_aff = math.log(aff_scale)
    sigma = F.softplus(spre) + min_scale
    y = ((action - aff_shift) / aff_scale).clamp(-1 + 1e-3, 1 - 1e-3)
    u = torch.atanh(y)
    z = (u - loc) / sigma
    log_det = 2.0 * (math.log(2.0) - u - F.softplus(-2.0 * u)) + log_aff
    logp_new = (-0.5 * z * z - torch.log(sigma) - 0.5 * math.log(2 * math.pi) - log_det).sum(-1)
    ratio = torch.exp(logp_new - old_logp)
    l1 = ratio * adv
    l2 = ratio.clamp(1 - clip_eps, 1 + clip_eps) * adv
    a_loss = -torch.minimum(l1, l2).mean()
    # MC entropy with externally-supplied standard normals (rsample)
    u_ent = loc + sigma * eps_ent
    log_det_e = 2.0 * (math.log(2.0) - u_ent - F.softplus(-2.0 * u_ent)) + log_aff
    logp_ent = (-0.5 * eps_ent * eps_ent - torch.log(sigma)
                - 0.5 * math.log(2 * math.pi) - log_det_e).sum(-1)
    entropy = (-logp_ent).mean()
    v_clip = old_value + (v_pred - old_value).clamp(-clip_eps, clip_eps)
    v_loss = 0.5 * torch.maximum((v_pred - targets) ** 2, (v_clip - targets) ** 2).mean()
    total = a_loss - ent_coef * entropy + vf_coef * v_loss
    return total, a_loss, v_loss, entropy


def closed_form_grads(loc, spre, v_pred, action, old_logp, old_value, adv,
                      targets, eps_ent, clip_eps, ent_coef, vf_coef,
                      min_scale, aff_scale, aff_shift):
    """Mirror of the kernel's analytic backward (mlp.hip
    ppo_head_loss_kernel)."""
    B = loc.shape[0]
    inv_B = 1.0 / B
    log_aff = math.log(aff_scale)
    sigma = F.softplus(spre) + min_scale
    y = ((action - aff_shift) / aff_scale).clamp(-1 + 1e-3, 1 - 1e-3)
    u = torch.atanh(y)
    z = (u - loc) / sigma
    log_det = 2.0 * (math.log(2.0) - u - F.softplus(-2.0 * u)) + log_aff
    logp_new = (-0.5 * z * z - torch.log(sigma) - 0.5 * math.log(2 * math.pi) - log_det).sum(-1)
    ratio = torch.exp(logp_new - old_logp)
    l1 = ratio * adv
    l2 = ratio.clamp(1 - clip_eps, 1 + clip_eps) * adv
    # d a_loss / d logp (kernel branch logic)
    unclipped = (ratio > 1 - clip_eps) & (ratio < 1 + clip_eps)
    dl_dlogp = torch.where(
        l1 <= l2, -ratio * adv,
        torch.where(unclipped, -ratio * adv, torch.zeros_like(ratio)),
    ) * inv_B
    dlogp_dloc = z / sigma
    dlogp_dsig = (z * z - 1.0) / sigma
    dloc = dl_dlogp[:, None] * dlogp_dloc
    dsig = dl_dlogp[:, None] * dlogp_dsig
    # entropy path
    u_ent = loc + sigma * eps_ent
    th = torch.tanh(u_ent)
    dent_dloc = -2.0 * th
    dent_dsig = 1.0 / sigma - 2.0 * th * eps_ent
    ce = -ent_coef * inv_B
    dloc = dloc + ce * dent_dloc
    dsig = dsig + ce * dent_dsig
    dspre = dsig * torch.sigmoid(spre)
    # value path
    v_clip = old_value + (v_pred - old_value).clamp(-clip_eps, clip_eps)
    e1 = v_pred - targets
    e2 = v_clip - targets
    dv = torch.where(
        e1 * e1 >= e2 * e2, e1,
        torch.where((v_pred - old_value).abs() < clip_eps, e2, torch.zeros_like(e1)),
    ) * (vf_coef * inv_B)
    return dloc, dspre, dv


@pytest.mark.parametrize("act_dim", [8, 4])
def test_head_loss_closed_form_matches_autograd(act_dim):
    B = 257
    dt = torch.float64
    loc = (torch.randn(B, act_dim, dtype=dt)).requires_grad_(True)
    spre = (torch.randn(B, act_dim, dtype=dt)).requires_grad_(True)
    v_pred = (torch.randn(B, dtype=dt)).requires_grad_(True)
    action = torch.tanh(torch.randn(B, act_dim, dtype=dt)) * 0.999
    old_logp = torch.randn(B, dtype=dt)
    old_value = torch.randn(B, dtype=dt)
    adv = torch.randn(B, dtype=dt)
    targets = torch.randn(B, dtype=dt)
    eps_ent = torch.randn(B, act_dim, dtype=dt)
    args = (action, old_logp, old_value, adv, targets, eps_ent,
            0.2, 0.01, 0.5, 1e-3, 1.0, 0.0)
    total, a_loss, v_loss, entropy = fused_head_reference(loc, spre, v_pred, *args)
    total.backward()
    dloc, dspre, dv = closed_form_grads(loc.detach(), spre.detach(),
                                        v_pred.detach(), *args)
    assert torch.allclose(loc.grad, dloc, atol=1e-10)
    assert torch.allclose(spre.grad, dspre, atol=1e-10)
    assert torch.allclose(v_pred.grad, dv, atol=1e-10)


def test_fused_reference_matches_eager_losses():
    """fused_head_reference's loss values == the eager path's losses
    (losses.py + AffineTanhTransformedDistribution)."""
    from stoix_amd.networks.distributions import AffineTanhTransformedDistribution
    from stoix_amd.ops.losses import clipped_value_loss, ppo_clip_loss

    B, A = 97, 8
    dt = torch.float64
    loc = torch.randn(B, A, dtype=dt)
    spre = torch.randn(B, A, dtype=dt)
    v_pred = torch.randn(B, dtype=dt)
    action = torch.tanh(torch.randn(B, A, dtype=dt)) * 0.98
    old_logp = torch.randn(B, dtype=dt)
    old_value = torch.randn(B, dtype=dt)
    adv = torch.randn(B, dtype=dt)
    targets = torch.randn(B, dtype=dt)
    eps_ent = torch.randn(B, A, dtype=dt)

    _, a_loss, v_loss, entropy = fused_head_reference(
        loc, spre, v_pred, action, old_logp, old_value, adv, targets,
        eps_ent, 0.2, 0.01, 0.5, 1e-3, 1.0, 0.0)

    sigma = F.softplus(spre) + 1e-3
    dist = AffineTanhTransformedDistribution(loc, sigma, -1.0, 1.0)
    logp = dist.log_prob(action)
    a_ref = ppo_clip_loss(logp, old_logp, adv, 0.2)
    v_ref = clipped_value_loss(v_pred, old_value, targets, 0.2)
    ent_ref = (-(dist._log_prob_from_u(loc + sigma * eps_ent))).mean()
    assert torch.allclose(a_loss, a_ref, atol=1e-9)
    assert torch.allclose(v_loss, v_ref, atol=1e-9)
    assert torch.allclose(entropy, ent_ref, atol=1e-9)


def test_chain_shared_buffer_w2_adjacency():
    """The fused engine's actor/critic chain layout: actor ends with W2,
    critic starts with W2, both flat16/grad16 slices of one buffer -> the
    [2, H, H] view over the boundary aliases exactly both W2 mirrors
    (zero-copy bmm operand) and one all-reduce covers both grads."""
    import torch

    from stoix_amd.systems.ppo.fused import _Chain

    H, K1P = 8, 32
    a_specs = [("W1", (H, K1P)), ("b1", (H,)), ("Wh", (16, H)), ("bh", (16,)),
               ("b2", (H,)), ("W2", (H, H))]
    c_specs = [("W2", (H, H)), ("W1", (H, K1P)), ("b1", (H,)), ("b2", (H,)),
               ("Wv", (H,)), ("bv", (1,))]
    nA = sum(int(torch.tensor(s).prod()) for _, s in a_specs)
    nC = sum(int(torch.tensor(s).prod()) for _, s in c_specs)
    big16 = torch.zeros(nA + nC, dtype=torch.bfloat16)
    biggrad = torch.zeros(nA + nC, dtype=torch.bfloat16)
    ac = _Chain(a_specs, "cpu", 1e-3, flat16=big16[:nA], grad16=biggrad[:nA])
    cc = _Chain(c_specs, "cpu", 1e-3, flat16=big16[nA:], grad16=biggrad[nA:])
    pair = big16[nA - H * H : nA + H * H].view(2, H, H)
    ac.views16["W2"].copy_(torch.full((H, H), 2.0, dtype=torch.bfloat16))
    cc.views16["W2"].copy_(torch.full((H, H), 3.0, dtype=torch.bfloat16))
    assert (pair[0] == 2.0).all() and (pair[1] == 3.0).all()
    # writing through the pair view is visible to the chains (same storage)
    pair[0, 0, 0] = 7.0
    assert float(ac.views16["W2"][0, 0]) == 7.0
    # one flat grad buffer covers both chains
    cc.gviews16["Wv"].fill_(1.0)
    assert float(biggrad.sum()) == H
