"""Batched MCTS correctness tests."""
import torch

from stoix_amd.search.mcts import mcts_search


def make_bandit_recurrent_fn(reward_per_action):
    """Deterministic 1-step bandit: action a yields reward_per_action[a],
    then the episode effectively ends (discount 0)."""
    rpa = reward_per_action

    def fn(embedding, action):
        B = action.shape[0]
        reward = rpa.to(action.device)[action]
        discount = torch.zeros(B)
        prior = torch.zeros(B, rpa.shape[0])
        value = torch.zeros(B)
        return dict(embedding), reward, discount, prior, value

    return fn


def test_mcts_finds_best_bandit_arm():
    B, A = 4, 3
    rewards = torch.tensor([0.1, 1.0, 0.3])
    fn = make_bandit_recurrent_fn(rewards)
    out = mcts_search(
        root_obs=torch.zeros(B, 2),
        root_embedding={"s": torch.zeros(B, 1)},
        root_prior_logits=torch.zeros(B, A),
        root_value=torch.zeros(B),
        recurrent_fn=fn,
        num_simulations=30,
        dirichlet_alpha=None,
        temperature=0.0,
        generator=torch.Generator().manual_seed(0),
    )
    assert (out.action == 1).all(), out.action_weights
    # best arm gets the majority of visits
    assert (out.action_weights[:, 1] > 0.5).all()
    # root value approaches the best reward as visits concentrate
    assert (out.search_value > 0.4).all()


def test_mcts_respects_prior_with_few_sims():
    """With strong priors and equal rewards, visits follow the prior."""
    B, A = 2, 4
    fn = make_bandit_recurrent_fn(torch.zeros(A))
    prior_logits = torch.tensor([[5.0, 0.0, 0.0, 0.0], [0.0, 0.0, 5.0, 0.0]])
    out = mcts_search(
        root_obs=torch.zeros(B, 2),
        root_embedding={"s": torch.zeros(B, 1)},
        root_prior_logits=prior_logits,
        root_value=torch.zeros(B),
        recurrent_fn=fn,
        num_simulations=20,
        dirichlet_alpha=None,
        temperature=0.0,
    )
    assert out.action[0] == 0 and out.action[1] == 2


def test_mcts_multistep_credit():
    """Two-step chain: action 0 leads to a state where reward 1 is available;
    action 1 gives 0 now and nothing later. Search should prefer action 0."""
    A = 2

    def fn(embedding, action):
        depth = embedding["d"]
        # at depth 0: no reward; at depth >= 1 reward only if first action was 0
        took0 = embedding["took0"]
        new_took0 = torch.where(depth.squeeze(-1) == 0, (action == 0).float(), took0.squeeze(-1)).unsqueeze(-1)
        reward = torch.where((depth.squeeze(-1) >= 1) & (new_took0.squeeze(-1) > 0.5),
                             torch.ones_like(depth.squeeze(-1)), torch.zeros_like(depth.squeeze(-1)))
        new_emb = {"d": depth + 1, "took0": new_took0}
        discount = torch.full((action.shape[0],), 0.95)
        prior = torch.zeros(action.shape[0], A)
        value = torch.zeros(action.shape[0])
        return new_emb, reward, discount, prior, value

    out = mcts_search(
        root_obs=torch.zeros(3, 2),
        root_embedding={"d": torch.zeros(3, 1), "took0": torch.zeros(3, 1)},
        root_prior_logits=torch.zeros(3, A),
        root_value=torch.zeros(3),
        recurrent_fn=fn,
        num_simulations=40,
        dirichlet_alpha=None,
        temperature=0.0,
    )
    assert (out.action == 0).all(), out.action_weights


def test_sampled_mcts_prefers_rewarding_arm():
    """Continuous sampled search: candidates near +1 yield reward; the
    search must concentrate visits on (and select) high-value candidates."""
    import torch
    from stoix_amd.search.mcts import sampled_mcts_search

    B, K, AD = 6, 6, 1
    torch.manual_seed(0)
    # candidates: fixed spread in [-1, 1]
    cand = torch.linspace(-1, 1, K).view(1, K, 1).repeat(B, 1, 1)

    def recurrent_fn(emb, action):
        # one-step bandit: reward = action value, episode ends (discount 0)
        # so arm quality is exactly the immediate reward
        r = action.squeeze(-1)
        new_cand = torch.linspace(-1, 1, K).view(1, K, 1).repeat(B, 1, 1)
        return emb, r, torch.zeros_like(r), new_cand, torch.zeros_like(r)

    out = sampled_mcts_search(
        root_obs=torch.zeros(B, 1),
        root_embedding={"h": torch.zeros(B, 1)},
        root_candidates=cand,
        root_value=torch.zeros(B),
        recurrent_fn=recurrent_fn,
        num_simulations=48,
        temperature=0.0,
    )
    assert out.action_weights.shape == (B, K)
    # best arm (action +1) must dominate the visit distribution
    assert (out.action_weights.argmax(-1) == K - 1).float().mean() > 0.8
    assert (out.action.squeeze(-1) > 0.5).float().mean() > 0.8


def test_mcts_backup_value_is_exact_on_deterministic_chain():
    """Value-backup precision on a SELF-CONSISTENT deterministic model:
    root arm 0 gives r=0.5, d=0.9 into a state worth exactly 2.0 (its own
    expansion pays 2.0 then terminates, agreeing with the value net), so
    EVERY simulation through arm 0 backs up exactly Q = 0.5 + 0.9*2.0 =
    2.3 at any depth; arm 1 is worth exactly 0. The root search value
    must converge to the visit-weighted mix and never exceed 2.3."""
    A = 2

    def fn(embedding, action):
        B = action.shape[0]
        depth = embedding["d"].squeeze(-1)
        took0 = embedding["took0"].squeeze(-1)
        at_root = depth == 0
        new_took0 = torch.where(at_root, (action == 0).float(), took0)
        reward = torch.where(
            at_root,
            torch.where(action == 0, torch.full((B,), 0.5), torch.zeros(B)),
            torch.where(new_took0 > 0.5, torch.full((B,), 2.0), torch.zeros(B)),
        )
        discount = torch.where(at_root, torch.full((B,), 0.9), torch.zeros(B))
        value = torch.where(
            at_root & (action == 0), torch.full((B,), 2.0), torch.zeros(B)
        )
        new_emb = {"d": embedding["d"] + 1, "took0": new_took0.unsqueeze(-1)}
        return new_emb, reward, discount, torch.zeros(B, A), value

    out = mcts_search(
        root_obs=torch.zeros(4, 2),
        root_embedding={"d": torch.zeros(4, 1), "took0": torch.zeros(4, 1)},
        root_prior_logits=torch.zeros(4, A),
        root_value=torch.zeros(4),
        recurrent_fn=fn,
        num_simulations=60,
        dirichlet_alpha=None,
        temperature=0.0,
    )
    assert (out.action == 0).all()
    assert (out.search_value <= 2.3 + 1e-5).all()
    # visits concentrate on arm 0 -> the mixed value sits well above half
    assert (out.search_value > 1.6).all(), out.search_value
