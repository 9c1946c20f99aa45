"""PopJym-class POMDP envs: velocity-masked classic control with the
start-flag + previous-action observation augmentation.

Restores the capability class of the reference's popjym suite
(/root/reference/stoix/utils/make_env.py:363-364 wraps popjym envs in
``AddStartFlagAndPrevAction``): partially observable tasks where the
optimal policy NEEDS memory — a feed-forward policy caps out, a recurrent
one solves them. The canonical instance is StatelessCartPole (velocities
masked out of the observation), the standard POMDP-ification used across
the POPGym line.

Observation layout: [masked_obs..., start_flag, prev_action_onehot...] —
the reference wrapper's exact augmentation semantics.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.classic import CartPole, Pendulum
from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace


class _MaskedObsEnv(StatefulVecEnv):
    """Base: wrap a fully-observable env class, keep only OBS_KEEP indices
    of its observation, append start flag + previous-action one-hot."""

    INNER_CLS = None
    OBS_KEEP: Tuple[int, ...] = ()

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self._inner = self.INNER_CLS(num_envs, device=device, seed=seed, **kw)
        # disable the inner HIP fast path: the wrapper drives _step_fn
        # directly (functional), so the augmentation composes on both CPU
        # and GPU torch paths
        self._inner._hip = None
        self.max_episode_steps = self._inner.max_episode_steps
        self.action_space = self._inner.action_space
        self._n_act = getattr(self.action_space, "num_values", 0) or 0
        base_dim = len(self.OBS_KEEP)
        self.observation_space = BoxSpace((base_dim + 1 + self._n_act,), -5.0, 5.0)
        self._keep = torch.tensor(self.OBS_KEEP, device=self.device)

    def _reset_fn(self, n: int) -> State:
        inner = self._inner._reset_fn(n)
        return {
            **inner,
            "_prev_a": torch.zeros(n, dtype=torch.long, device=self.device),
            "_is_start": torch.ones(n, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        full = self._inner._obs_fn(inner_state)
        masked = full.index_select(-1, self._keep)
        onehot = torch.nn.functional.one_hot(
            state["_prev_a"].clamp(0, max(self._n_act - 1, 0)), max(self._n_act, 1)
        ).float()
        return torch.cat([masked, state["_is_start"].unsqueeze(-1), onehot], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        new_inner, reward, terminated = self._inner._step_fn(inner_state, action)
        return (
            {
                **new_inner,
                "_prev_a": action.long(),
                "_is_start": torch.zeros_like(state["_is_start"]),
            },
            reward,
            terminated,
        )


class StatelessCartPole(_MaskedObsEnv):
    """CartPole with the two velocity components hidden (obs = [x, theta]
    + flag + prev action): the POPGym StatelessCartPole task."""

    INNER_CLS = CartPole
    OBS_KEEP = (0, 2)  # x, theta (drop x_dot, theta_dot)
    max_episode_steps = 500
    solved_return_threshold = 450.0


class NoisyStatelessCartPole(StatelessCartPole):
    """StatelessCartPole with observation noise (POPGym 'noisy' variant)."""

    NOISE = 0.1

    def _obs_fn(self, state: State) -> Tensor:
        obs = super()._obs_fn(state)
        n = obs.shape[0]
        noise = torch.randn(n, 2, device=self.device, generator=self.gen) * self.NOISE
        obs = obs.clone()
        obs[:, :2] = obs[:, :2] + noise
        return obs


class StatelessPendulum(_MaskedObsEnv):
    """Pendulum with angular velocity hidden (obs = [cos, sin] + flag +
    prev-action placeholder; continuous action -> no one-hot)."""

    INNER_CLS = Pendulum
    OBS_KEEP = (0, 1)  # cos(theta), sin(theta); drop theta_dot
    max_episode_steps = 200

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed, **kw)
        base_dim = len(self.OBS_KEEP)
        # continuous action: append the raw previous action instead
        adim = self.action_space.shape[0]
        self.observation_space = BoxSpace((base_dim + 1 + adim,), -5.0, 5.0)

    def _reset_fn(self, n: int) -> State:
        inner = self._inner._reset_fn(n)
        adim = self.action_space.shape[0]
        return {
            **inner,
            "_prev_a": torch.zeros(n, adim, device=self.device),
            "_is_start": torch.ones(n, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        full = self._inner._obs_fn(inner_state)
        masked = full.index_select(-1, self._keep)
        return torch.cat(
            [masked, state["_is_start"].unsqueeze(-1), state["_prev_a"]], dim=-1
        )

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        new_inner, reward, terminated = self._inner._step_fn(inner_state, action)
        return (
            {
                **new_inner,
                "_prev_a": action.float().reshape(state["_prev_a"].shape),
                "_is_start": torch.zeros_like(state["_is_start"]),
            },
            reward,
            terminated,
        )


class _MemoryGameEnv(StatefulVecEnv):
    """Base for POPGym-class pure memory games (AutoEncode / CountRecall /
    RepeatFirst — the reference's popjym scenario families beyond the
    stateless-control tasks, configs/env/popjym/*.yaml). Original vectorised
    designs of the same capability class: episodic symbol tasks where the
    score is the fraction of correct recalls, rewards ±1/n_scored (POPGym's
    normalised-return convention), and the observation carries the
    reference wrapper's start-flag + previous-action augmentation
    (make_env.py:363-364)."""

    capture_safe = True

    def _prev_a_onehot(self, state: State) -> Tensor:
        n_act = self.action_space.num_values
        return torch.nn.functional.one_hot(
            state["_prev_a"].clamp(0, n_act - 1), n_act
        ).float() * (state["t"] > 0).unsqueeze(-1).float()


class RepeatFirst(_MemoryGameEnv):
    """Observe a symbol at t=0; repeat it at every later step.

    Obs: [symbol one-hot (zeros after t=0), start flag, time fraction,
    prev-action one-hot]. Reward ±1/(L-1) per answer step.
    """

    L = 16
    A = 4
    solved_return_threshold = 0.8

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.max_episode_steps = self.L + 1
        self.action_space = DiscreteSpace(self.A)
        self.observation_space = BoxSpace((self.A + 2 + self.A,), 0.0, 1.0)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        return {
            "target": torch.randint(0, self.A, (n,), device=dev, generator=self.gen),
            "t": torch.zeros(n, dtype=torch.long, device=dev),
            "_prev_a": torch.zeros(n, dtype=torch.long, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        show = (state["t"] == 0).unsqueeze(-1).float()
        sym = torch.nn.functional.one_hot(state["target"], self.A).float() * show
        tf = (state["t"].float() / self.L).unsqueeze(-1)
        return torch.cat([sym, show, tf, self._prev_a_onehot(state)], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        a = action.long().clamp(0, self.A - 1)
        scored = state["t"] >= 1
        correct = (a == state["target"]) & scored
        reward = (correct.float() - (scored & ~correct).float()) / float(self.L - 1)
        t = state["t"] + 1
        return (
            {"target": state["target"], "t": t, "_prev_a": a},
            reward,
            t >= self.L,
        )


class RepeatFirstEasy(RepeatFirst):
    L, A = 16, 4


class RepeatFirstMedium(RepeatFirst):
    L, A = 48, 8


class RepeatFirstHard(RepeatFirst):
    L, A = 104, 16


class AutoEncode(_MemoryGameEnv):
    """Watch a symbol sequence (t = 0..L-1), then reproduce it in order
    during the play phase (t = L..2L-1).

    Obs: [symbol one-hot (watch phase only), play-phase flag, start flag,
    time fraction, prev-action one-hot]. Reward ±1/L per play answer; the
    play-step obs carries no symbol, so the whole sequence must be held in
    memory.
    """

    L = 6
    A = 4
    solved_return_threshold = 0.8

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.max_episode_steps = 2 * self.L + 1
        self.action_space = DiscreteSpace(self.A)
        self.observation_space = BoxSpace((self.A + 3 + self.A,), 0.0, 1.0)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        return {
            "seq": torch.randint(0, self.A, (n, self.L), device=dev, generator=self.gen),
            "t": torch.zeros(n, dtype=torch.long, device=dev),
            "_prev_a": torch.zeros(n, dtype=torch.long, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        t = state["t"]
        watch = (t < self.L).unsqueeze(-1).float()
        idx = t.clamp(max=self.L - 1)
        cur = state["seq"].gather(1, idx.unsqueeze(1)).squeeze(1)
        sym = torch.nn.functional.one_hot(cur, self.A).float() * watch
        play = 1.0 - watch
        start = (t == 0).unsqueeze(-1).float()
        tf = (t.float() / (2 * self.L)).unsqueeze(-1)
        return torch.cat([sym, play, start, tf, self._prev_a_onehot(state)], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        a = action.long().clamp(0, self.A - 1)
        t = state["t"]
        scored = t >= self.L
        ans_idx = (t - self.L).clamp(0, self.L - 1)
        expect = state["seq"].gather(1, ans_idx.unsqueeze(1)).squeeze(1)
        correct = (a == expect) & scored
        reward = (correct.float() - (scored & ~correct).float()) / float(self.L)
        t = t + 1
        return (
            {"seq": state["seq"], "t": t, "_prev_a": a},
            reward,
            t >= 2 * self.L,
        )


class AutoEncodeEasy(AutoEncode):
    L, A = 6, 4


class AutoEncodeMedium(AutoEncode):
    L, A = 12, 6


class CountRecall(_MemoryGameEnv):
    """Each step shows a value symbol and a query symbol; answer how many
    times the query value has occurred so far (including this step's
    value). Running counts must be maintained in memory.

    Obs: [value one-hot, query one-hot, start flag, time fraction,
    prev-action one-hot]. Action space Discrete(T+1) (the count). Reward
    ±1/T per answer.
    """

    T = 16
    V = 4
    solved_return_threshold = 0.6

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.max_episode_steps = self.T + 1
        self.action_space = DiscreteSpace(self.T + 1)
        self.observation_space = BoxSpace(
            (2 * self.V + 2 + self.T + 1,), 0.0, 1.0
        )

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        s0 = torch.randint(0, self.V, (n,), device=dev, generator=self.gen)
        q0 = torch.randint(0, self.V, (n,), device=dev, generator=self.gen)
        counts = torch.nn.functional.one_hot(s0, self.V).long()
        return {
            "counts": counts,
            "s": s0,
            "q": q0,
            "t": torch.zeros(n, dtype=torch.long, device=dev),
            "_prev_a": torch.zeros(n, dtype=torch.long, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        sv = torch.nn.functional.one_hot(state["s"], self.V).float()
        qv = torch.nn.functional.one_hot(state["q"], self.V).float()
        start = (state["t"] == 0).unsqueeze(-1).float()
        tf = (state["t"].float() / self.T).unsqueeze(-1)
        return torch.cat([sv, qv, start, tf, self._prev_a_onehot(state)], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        a = action.long().clamp(0, self.T)
        truth = state["counts"].gather(1, state["q"].unsqueeze(1)).squeeze(1)
        correct = a == truth.clamp(max=self.T)
        reward = (correct.float() - (~correct).float()) / float(self.T)
        dev = self.device
        n = a.shape[0]
        s = torch.randint(0, self.V, (n,), device=dev, generator=self.gen)
        q = torch.randint(0, self.V, (n,), device=dev, generator=self.gen)
        counts = state["counts"] + torch.nn.functional.one_hot(s, self.V).long()
        t = state["t"] + 1
        return (
            {"counts": counts, "s": s, "q": q, "t": t, "_prev_a": a},
            reward,
            t >= self.T,
        )


class CountRecallEasy(CountRecall):
    T, V = 16, 4


class CountRecallMedium(CountRecall):
    T, V = 32, 8
