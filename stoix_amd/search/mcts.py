"""Batched array-based MCTS on device (the mctx replacement).

Functional parity with the mctx ``muzero_policy`` / ``gumbel_muzero_policy``
usage in the reference (/root/reference/stoix/systems/search/ff_az.py:377-379,
ff_mz.py): a fixed number of simulations over a preallocated node arena
([B, S+1] struct-of-arrays tensors), PUCT selection with min-max value
normalisation, expansion through a user ``recurrent_fn`` (real env step for
AlphaZero, learned dynamics for MuZero), and masked backward passes. Every
phase is a batched tensor op over all B trees simultaneously — the MI355X
execution shape for K16 of SURVEY.md §2.9.

``recurrent_fn(embedding, action) -> (embedding, reward, discount, prior_logits,
value)`` where ``embedding`` is a dict of [B, ...] tensors.
"""
from __future__ import annotations

from typing import Callable, Dict, NamedTuple, Optional, Tuple

import torch

Tensor = torch.Tensor


class SearchOutput(NamedTuple):
    action: Tensor  # [B] chosen action
    action_weights: Tensor  # [B, A] normalised root visit counts
    search_value: Tensor  # [B] root value estimate


class _Arena:
    def __init__(self, B: int, N: int, A: int, device, value0: Tensor):
        z = lambda *s, dtype=torch.float32: torch.zeros(*s, dtype=dtype, device=device)
        self.visit = z(B, N)
        self.value_sum = z(B, N)
        self.reward = z(B, N)
        self.discount = z(B, N)
        self.prior = z(B, N, A)
        self.children = torch.full((B, N, A), -1, dtype=torch.long, device=device)
        self.parent = torch.full((B, N), -1, dtype=torch.long, device=device)
        self.act_from_parent = z(B, N, dtype=torch.long)
        # min-max value normalisation bounds per tree
        self.min_q = torch.full((B,), 1e9, device=device)
        self.max_q = torch.full((B,), -1e9, device=device)
        self.batch = torch.arange(B, device=device)

    def q_value(self, nodes: Tensor) -> Tensor:
        v = self.visit[self.batch, nodes]
        return torch.where(v > 0, self.value_sum[self.batch, nodes] / v.clamp(min=1), torch.zeros_like(v))


def mcts_search(
    root_obs: Tensor,
    root_embedding: Dict[str, Tensor],
    root_prior_logits: Tensor,
    root_value: Tensor,
    recurrent_fn: Callable,
    num_simulations: int,
    c_puct: float = 1.25,
    dirichlet_alpha: Optional[float] = 0.3,
    dirichlet_fraction: float = 0.25,
    temperature: float = 1.0,
    generator: Optional[torch.Generator] = None,
    gumbel: bool = False,
) -> SearchOutput:
    device = root_obs.device
    B, A = root_prior_logits.shape
    N = num_simulations + 1
    arena = _Arena(B, N, A, device, root_value)
    bidx = arena.batch

    prior = torch.softmax(root_prior_logits, dim=-1)
    if dirichlet_alpha is not None and dirichlet_fraction > 0:
        noise = torch._standard_gamma(
            torch.full((B, A), dirichlet_alpha, device=device)
        )
        noise = noise / noise.sum(-1, keepdim=True).clamp(min=1e-9)
        prior = (1 - dirichlet_fraction) * prior + dirichlet_fraction * noise
    arena.prior[:, 0] = prior
    arena.discount[:, 0] = 1.0

    # embedding arena: [B, N, ...] per field
    emb_arena = {
        k: torch.zeros((B, N, *v.shape[1:]), dtype=v.dtype, device=device)
        for k, v in root_embedding.items()
    }
    for k, v in root_embedding.items():
        emb_arena[k][:, 0] = v

    # seed the root with its network value
    arena.visit[:, 0] = 1.0
    arena.value_sum[:, 0] = root_value
    arena.min_q = torch.minimum(arena.min_q, root_value)
    arena.max_q = torch.maximum(arena.max_q, root_value)

    for sim in range(1, num_simulations + 1):
        # ---- selection: descend PUCT until a missing child edge
        node = torch.zeros(B, dtype=torch.long, device=device)
        descending = torch.ones(B, dtype=torch.bool, device=device)
        sel_parent = torch.zeros(B, dtype=torch.long, device=device)
        sel_action = torch.zeros(B, dtype=torch.long, device=device)
        for _ in range(sim):
            child_idx = arena.children[bidx, node]  # [B, A]
            child_visit = torch.where(
                child_idx >= 0, arena.visit[bidx.unsqueeze(1), child_idx.clamp(min=0)], torch.zeros_like(child_idx, dtype=torch.float32)
            )
            child_q = torch.where(
                child_idx >= 0,
                arena.value_sum[bidx.unsqueeze(1), child_idx.clamp(min=0)] / child_visit.clamp(min=1),
                torch.zeros_like(child_visit),
            )
            # child value from the parent's perspective: r + gamma*q, min-max normalised
            child_r = torch.where(
                child_idx >= 0, arena.reward[bidx.unsqueeze(1), child_idx.clamp(min=0)], torch.zeros_like(child_visit)
            )
            child_g = torch.where(
                child_idx >= 0, arena.discount[bidx.unsqueeze(1), child_idx.clamp(min=0)], torch.zeros_like(child_visit)
            )
            q_edge = child_r + child_g * child_q
            span = (arena.max_q - arena.min_q).clamp(min=1e-3).unsqueeze(1)
            # unvisited children score as the PARENT's normalised value
            # (mctx qtransform_by_parent_and_siblings): scoring them as the
            # global minimum lets one mediocre visited arm starve its
            # unvisited siblings for hundreds of simulations
            parent_q = (arena.q_value(node) - arena.min_q) / span.squeeze(1)
            q_norm = torch.where(
                child_visit > 0,
                (q_edge - arena.min_q.unsqueeze(1)) / span,
                parent_q.unsqueeze(1).expand_as(q_edge),
            )
            parent_visit = arena.visit[bidx, node].unsqueeze(1)
            ucb = q_norm + c_puct * arena.prior[bidx, node] * torch.sqrt(parent_visit.clamp(min=1)) / (
                1.0 + child_visit
            )
            best_a = ucb.argmax(dim=-1)
            has_child = child_idx.gather(1, best_a.unsqueeze(1)).squeeze(1) >= 0
            # record the (parent, action) where the descent stops
            stop_here = descending & ~has_child
            sel_parent = torch.where(stop_here, node, sel_parent)
            sel_action = torch.where(stop_here, best_a, sel_action)
            step_child = child_idx.gather(1, best_a.unsqueeze(1)).squeeze(1).clamp(min=0)
            node = torch.where(descending & has_child, step_child, node)
            descending = descending & has_child
        # trees still descending at max depth expand from their current node
        sel_parent = torch.where(descending, node, sel_parent)
        if bool(descending.any()):
            # pick their PUCT action at the final node (recompute quickly)
            child_idx = arena.children[bidx, node]
            ucb = arena.prior[bidx, node]
            best_a = ucb.argmax(dim=-1)
            sel_action = torch.where(descending, best_a, sel_action)

        # ---- expansion: recurrent_fn on the selected (parent, action)
        parent_emb = {k: v[bidx, sel_parent] for k, v in emb_arena.items()}
        new_emb, reward, discount, prior_logits, value = recurrent_fn(parent_emb, sel_action)
        new_node = sim
        for k, v in new_emb.items():
            emb_arena[k][:, new_node] = v
        arena.reward[:, new_node] = reward
        arena.discount[:, new_node] = discount
        arena.prior[:, new_node] = torch.softmax(prior_logits, dim=-1)
        arena.parent[:, new_node] = sel_parent
        arena.act_from_parent[:, new_node] = sel_action
        arena.children[bidx, sel_parent, sel_action] = new_node

        # ---- backward: propagate value to the root (masked walk)
        g = value.clone()
        cur = torch.full((B,), new_node, dtype=torch.long, device=device)
        alive = torch.ones(B, dtype=torch.bool, device=device)
        for _ in range(sim + 1):
            af = alive.to(torch.float32)
            arena.visit[bidx, cur] += af
            arena.value_sum[bidx, cur] += g * af
            q_here = arena.value_sum[bidx, cur] / arena.visit[bidx, cur].clamp(min=1)
            arena.min_q = torch.where(alive, torch.minimum(arena.min_q, q_here), arena.min_q)
            arena.max_q = torch.where(alive, torch.maximum(arena.max_q, q_here), arena.max_q)
            g = arena.reward[bidx, cur] + arena.discount[bidx, cur] * g
            nxt = arena.parent[bidx, cur]
            alive = alive & (nxt >= 0)
            cur = nxt.clamp(min=0)

    # ---- readout
    root_children = arena.children[:, 0]  # [B, A]
    counts = torch.where(
        root_children >= 0,
        arena.visit[bidx.unsqueeze(1), root_children.clamp(min=0)],
        torch.zeros_like(root_children, dtype=torch.float32),
    )
    weights = counts / counts.sum(-1, keepdim=True).clamp(min=1e-9)
    search_value = arena.value_sum[:, 0] / arena.visit[:, 0].clamp(min=1)
    if temperature <= 0:
        action = weights.argmax(dim=-1)
    else:
        logits = torch.log(weights.clamp(min=1e-9)) / temperature
        u = torch.rand(logits.shape, device=device, generator=generator)
        gnoise = -torch.log((-torch.log(u.clamp(min=1e-10))).clamp(min=1e-10))
        action = (logits + gnoise).argmax(dim=-1)
    return SearchOutput(action=action, action_weights=weights, search_value=search_value)


class SampledSearchOutput(NamedTuple):
    action: Tensor  # [B, act_dim] chosen continuous action
    sampled_actions: Tensor  # [B, K, act_dim] root candidate actions
    action_weights: Tensor  # [B, K] normalised root visit counts
    search_value: Tensor  # [B] root value estimate


def sampled_mcts_search(
    root_obs: Tensor,
    root_embedding: Dict[str, Tensor],
    root_candidates: Tensor,  # [B, K, act_dim] sampled from the root policy
    root_value: Tensor,
    recurrent_fn: Callable,
    num_simulations: int,
    c_puct: float = 1.25,
    temperature: float = 1.0,
    generator: Optional[torch.Generator] = None,
) -> SampledSearchOutput:
    """Sampled MCTS for continuous actions (parity surface: mctx's sampled
    policies as used by /root/reference/stoix/systems/search/ff_sampled_az.py
    and ff_sampled_mz.py).

    Each node holds K candidate actions drawn from the node's policy; the
    tree then searches over the K discrete arms. Because arms are i.i.d.
    samples from the current policy, the sampled prior over arms is uniform
    (the policy density is already represented by the sampling itself —
    the Sampled MuZero beta-correction cancels for beta = pi).

    ``recurrent_fn(embedding, action[B, act_dim]) ->
        (embedding, reward, discount, candidates[B, K, act_dim], value)``
    """
    device = root_obs.device
    B, K, act_dim = root_candidates.shape
    N = num_simulations + 1
    arena = _Arena(B, N, K, device, root_value)
    bidx = arena.batch
    arena.prior[:, 0] = 1.0 / K
    arena.discount[:, 0] = 1.0
    cand = torch.zeros((B, N, K, act_dim), device=device)
    cand[:, 0] = root_candidates

    emb_arena = {
        k: torch.zeros((B, N, *v.shape[1:]), dtype=v.dtype, device=device)
        for k, v in root_embedding.items()
    }
    for k, v in root_embedding.items():
        emb_arena[k][:, 0] = v

    arena.visit[:, 0] = 1.0
    arena.value_sum[:, 0] = root_value
    arena.min_q = torch.minimum(arena.min_q, root_value)
    arena.max_q = torch.maximum(arena.max_q, root_value)

    for sim in range(1, num_simulations + 1):
        node = torch.zeros(B, dtype=torch.long, device=device)
        descending = torch.ones(B, dtype=torch.bool, device=device)
        sel_parent = torch.zeros(B, dtype=torch.long, device=device)
        sel_action = torch.zeros(B, dtype=torch.long, device=device)
        for _ in range(sim):
            child_idx = arena.children[bidx, node]
            child_visit = torch.where(
                child_idx >= 0,
                arena.visit[bidx.unsqueeze(1), child_idx.clamp(min=0)],
                torch.zeros_like(child_idx, dtype=torch.float32),
            )
            child_q = torch.where(
                child_idx >= 0,
                arena.value_sum[bidx.unsqueeze(1), child_idx.clamp(min=0)] / child_visit.clamp(min=1),
                torch.zeros_like(child_visit),
            )
            child_r = torch.where(
                child_idx >= 0,
                arena.reward[bidx.unsqueeze(1), child_idx.clamp(min=0)],
                torch.zeros_like(child_visit),
            )
            child_g = torch.where(
                child_idx >= 0,
                arena.discount[bidx.unsqueeze(1), child_idx.clamp(min=0)],
                torch.zeros_like(child_visit),
            )
            q_edge = child_r + child_g * child_q
            span = (arena.max_q - arena.min_q).clamp(min=1e-3).unsqueeze(1)
            # unvisited children score as the PARENT's normalised value
            # (mctx qtransform_by_parent_and_siblings): scoring them as the
            # global minimum lets one mediocre visited arm starve its
            # unvisited siblings for hundreds of simulations
            parent_q = (arena.q_value(node) - arena.min_q) / span.squeeze(1)
            q_norm = torch.where(
                child_visit > 0,
                (q_edge - arena.min_q.unsqueeze(1)) / span,
                parent_q.unsqueeze(1).expand_as(q_edge),
            )
            parent_visit = arena.visit[bidx, node].unsqueeze(1)
            ucb = q_norm + c_puct * arena.prior[bidx, node] * torch.sqrt(
                parent_visit.clamp(min=1)
            ) / (1.0 + child_visit)
            best_a = ucb.argmax(dim=-1)
            has_child = child_idx.gather(1, best_a.unsqueeze(1)).squeeze(1) >= 0
            stop_here = descending & ~has_child
            sel_parent = torch.where(stop_here, node, sel_parent)
            sel_action = torch.where(stop_here, best_a, sel_action)
            step_child = child_idx.gather(1, best_a.unsqueeze(1)).squeeze(1).clamp(min=0)
            node = torch.where(descending & has_child, step_child, node)
            descending = descending & has_child
        sel_parent = torch.where(descending, node, sel_parent)
        if bool(descending.any()):
            best_a = arena.prior[bidx, node].argmax(dim=-1)
            sel_action = torch.where(descending, best_a, sel_action)

        parent_emb = {k: v[bidx, sel_parent] for k, v in emb_arena.items()}
        action_cont = cand[bidx, sel_parent, sel_action]  # [B, act_dim]
        new_emb, reward, discount, new_cand, value = recurrent_fn(parent_emb, action_cont)
        new_node = sim
        for k, v in new_emb.items():
            emb_arena[k][:, new_node] = v
        arena.reward[:, new_node] = reward
        arena.discount[:, new_node] = discount
        arena.prior[:, new_node] = 1.0 / K
        cand[:, new_node] = new_cand
        arena.parent[:, new_node] = sel_parent
        arena.act_from_parent[:, new_node] = sel_action
        arena.children[bidx, sel_parent, sel_action] = new_node

        g = value.clone()
        cur = torch.full((B,), new_node, dtype=torch.long, device=device)
        alive = torch.ones(B, dtype=torch.bool, device=device)
        for _ in range(sim + 1):
            af = alive.to(torch.float32)
            arena.visit[bidx, cur] += af
            arena.value_sum[bidx, cur] += g * af
            q_here = arena.value_sum[bidx, cur] / arena.visit[bidx, cur].clamp(min=1)
            arena.min_q = torch.where(alive, torch.minimum(arena.min_q, q_here), arena.min_q)
            arena.max_q = torch.where(alive, torch.maximum(arena.max_q, q_here), arena.max_q)
            g = arena.reward[bidx, cur] + arena.discount[bidx, cur] * g
            nxt = arena.parent[bidx, cur]
            alive = alive & (nxt >= 0)
            cur = nxt.clamp(min=0)

    root_children = arena.children[:, 0]
    counts = torch.where(
        root_children >= 0,
        arena.visit[bidx.unsqueeze(1), root_children.clamp(min=0)],
        torch.zeros_like(root_children, dtype=torch.float32),
    )
    weights = counts / counts.sum(-1, keepdim=True).clamp(min=1e-9)
    search_value = arena.value_sum[:, 0] / arena.visit[:, 0].clamp(min=1)
    if temperature <= 0:
        arm = weights.argmax(dim=-1)
    else:
        logits = torch.log(weights.clamp(min=1e-9)) / temperature
        u = torch.rand(logits.shape, device=device, generator=generator)
        gnoise = -torch.log((-torch.log(u.clamp(min=1e-10))).clamp(min=1e-10))
        arm = (logits + gnoise).argmax(dim=-1)
    action = root_candidates[bidx, arm]
    return SampledSearchOutput(
        action=action,
        sampled_actions=root_candidates,
        action_weights=weights,
        search_value=search_value,
    )
