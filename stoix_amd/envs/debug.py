"""Diagnostic debug environments.

Same purpose as the reference's five debug games
(/root/reference/stoix/utils/debug_env.py:25-411: identity, sequence,
delayed_reward, discount_sensitive, exploration): tiny deterministic games
with known optimal behaviour, used as learning-sanity fixtures. The games
here are original designs serving the same diagnostic roles.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace


class IdentityGame(StatefulVecEnv):
    """Obs is a one-hot symbol; reward 1 iff action == symbol. Tests basic
    policy learning. Optimal return = episode length."""

    def __init__(self, num_envs, device="cpu", seed=0, num_symbols: int = 4, episode_length: int = 10, **kw):
        super().__init__(num_envs, device, seed)
        self.num_symbols = num_symbols
        self.max_episode_steps = episode_length
        self.solved_return_threshold = float(episode_length) * 0.95
        self.observation_space = BoxSpace((num_symbols,), 0.0, 1.0)
        self.action_space = DiscreteSpace(num_symbols)

    def _reset_fn(self, n: int) -> State:
        return {"sym": self.randint(self.num_symbols, n)}

    def _obs_fn(self, state: State) -> Tensor:
        return torch.nn.functional.one_hot(state["sym"], self.num_symbols).to(torch.float32)

    def _step_fn(self, state, action) -> Tuple[State, Tensor, Tensor]:
        n = action.shape[0]
        reward = (action.long() == state["sym"]).to(torch.float32)
        new_sym = self.randint(self.num_symbols, n)
        terminated = torch.zeros(n, dtype=torch.bool, device=self.device)
        return {"sym": new_sym}, reward, terminated


class SequenceGame(StatefulVecEnv):
    """The target symbol is only visible at step 0; reward at the final step
    iff the final action equals the remembered symbol. Tests memory (RNN)."""

    def __init__(self, num_envs, device="cpu", seed=0, num_symbols: int = 4, episode_length: int = 6, **kw):
        super().__init__(num_envs, device, seed)
        self.num_symbols = num_symbols
        self.max_episode_steps = episode_length
        self.solved_return_threshold = 0.95
        self.observation_space = BoxSpace((num_symbols + 1,), 0.0, 1.0)
        self.action_space = DiscreteSpace(num_symbols)

    def _reset_fn(self, n: int) -> State:
        return {
            "sym": self.randint(self.num_symbols, n),
            "t": torch.zeros(n, dtype=torch.long, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        one_hot = torch.nn.functional.one_hot(state["sym"], self.num_symbols).to(torch.float32)
        visible = (state["t"] == 0).to(torch.float32).unsqueeze(-1)
        t_frac = (state["t"].to(torch.float32) / self.max_episode_steps).unsqueeze(-1)
        return torch.cat([one_hot * visible, t_frac], dim=-1)

    def _step_fn(self, state, action) -> Tuple[State, Tensor, Tensor]:
        n = action.shape[0]
        t = state["t"] + 1
        is_final = t >= self.max_episode_steps
        reward = torch.where(
            is_final & (action.long() == state["sym"]),
            torch.ones(n, device=self.device),
            torch.zeros(n, device=self.device),
        )
        terminated = torch.zeros(n, dtype=torch.bool, device=self.device)
        return {"sym": state["sym"], "t": t}, reward, terminated


class DelayedRewardGame(StatefulVecEnv):
    """The step-0 action decides a reward paid only at the final step.
    Tests credit assignment over a delay."""

    def __init__(self, num_envs, device="cpu", seed=0, episode_length: int = 8, **kw):
        super().__init__(num_envs, device, seed)
        self.max_episode_steps = episode_length
        self.solved_return_threshold = 0.95
        self.observation_space = BoxSpace((2,), 0.0, 1.0)
        self.action_space = DiscreteSpace(2)

    def _reset_fn(self, n: int) -> State:
        return {
            "t": torch.zeros(n, dtype=torch.long, device=self.device),
            "chose_good": torch.zeros(n, dtype=torch.bool, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        t_frac = state["t"].to(torch.float32) / self.max_episode_steps
        first = (state["t"] == 0).to(torch.float32)
        return torch.stack([first, t_frac], dim=-1)

    def _step_fn(self, state, action) -> Tuple[State, Tensor, Tensor]:
        t = state["t"]
        chose_good = torch.where(t == 0, action.long() == 1, state["chose_good"])
        t = t + 1
        is_final = t >= self.max_episode_steps
        n = action.shape[0]
        reward = torch.where(
            is_final, torch.where(chose_good, 1.0, -1.0), torch.zeros(n, device=self.device)
        )
        terminated = torch.zeros(n, dtype=torch.bool, device=self.device)
        return {"t": t, "chose_good": chose_good}, reward, terminated


class DiscountSensitiveGame(StatefulVecEnv):
    """Action 0 pays +0.6 immediately and ends; action 1 pays +1.0 after a
    delay. The optimal action flips with gamma — tests discounting."""

    def __init__(self, num_envs, device="cpu", seed=0, delay: int = 5, **kw):
        super().__init__(num_envs, device, seed)
        self.delay = delay
        self.max_episode_steps = delay + 2
        self.observation_space = BoxSpace((2,), 0.0, 1.0)
        self.action_space = DiscreteSpace(2)

    def _reset_fn(self, n: int) -> State:
        return {
            "t": torch.zeros(n, dtype=torch.long, device=self.device),
            "waiting": torch.zeros(n, dtype=torch.bool, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        return torch.stack(
            [(state["t"] == 0).to(torch.float32), state["waiting"].to(torch.float32)], dim=-1
        )

    def _step_fn(self, state, action) -> Tuple[State, Tensor, Tensor]:
        t = state["t"]
        first = t == 0
        waiting = torch.where(first, action.long() == 1, state["waiting"])
        t = t + 1
        took_now = first & (action.long() == 0)
        delayed_pay = waiting & (t >= self.delay)
        reward = torch.where(took_now, 0.6, torch.where(delayed_pay, 1.0, 0.0)).to(torch.float32)
        terminated = took_now | delayed_pay
        return {"t": t, "waiting": waiting}, reward, terminated


class ExplorationChain(StatefulVecEnv):
    """N-state chain: going left at state 0 pays 0.01 each step; reaching the
    far right end pays 1.0 and terminates. Greedy-myopic policies stay left —
    tests exploration."""

    def __init__(self, num_envs, device="cpu", seed=0, chain_length: int = 10, **kw):
        super().__init__(num_envs, device, seed)
        self.chain_length = chain_length
        self.max_episode_steps = 2 * chain_length
        self.solved_return_threshold = 0.99
        self.observation_space = BoxSpace((chain_length,), 0.0, 1.0)
        self.action_space = DiscreteSpace(2)

    def _reset_fn(self, n: int) -> State:
        return {"pos": torch.zeros(n, dtype=torch.long, device=self.device)}

    def _obs_fn(self, state: State) -> Tensor:
        return torch.nn.functional.one_hot(state["pos"], self.chain_length).to(torch.float32)

    def _step_fn(self, state, action) -> Tuple[State, Tensor, Tensor]:
        pos = state["pos"]
        right = action.long() == 1
        new_pos = torch.where(right, (pos + 1).clamp(max=self.chain_length - 1), (pos - 1).clamp(min=0))
        at_left = (pos == 0) & ~right
        reached_goal = new_pos == self.chain_length - 1
        reward = torch.where(reached_goal, 1.0, torch.where(at_left, 0.01, 0.0)).to(torch.float32)
        return {"pos": new_pos}, reward, reached_goal


DEBUG_ENVIRONMENTS = {
    "identity": IdentityGame,
    "sequence": SequenceGame,
    "delayed_reward": DelayedRewardGame,
    "discount_sensitive": DiscountSensitiveGame,
    "exploration": ExplorationChain,
}
