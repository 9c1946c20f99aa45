"""Anakin Recurrent PPO (parity: /root/reference/stoix/systems/ppo/anakin/
rec_ppo.py): ScannedRNN actor/critic with done-masked hidden resets,
transitions carry hidden states; the update re-runs the RNN from each
minibatch's initial stored hidden state over the full sequence (:210-250);
minibatches permute the ENV axis only, keeping time contiguous (:340-370);
bootstrap from the value trace's final value (:165-176).
"""
from __future__ import annotations

import sys
from typing import Any, Dict

import torch
import torch.nn as nn

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.factory import build_recurrent_actor, build_recurrent_critic
from stoix_amd.ops import multistep
from stoix_amd.ops.losses import clipped_value_loss, ppo_clip_loss
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


def _index_state(state: Any, idx: Tensor) -> Any:
    if isinstance(state, tuple):
        return tuple(_index_state(s, idx) for s in state)
    if isinstance(state, list):
        return [_index_state(s, idx) for s in state]
    return state[idx]


def _clone_state(state: Any) -> Any:
    if isinstance(state, tuple):
        return tuple(_clone_state(s) for s in state)
    if isinstance(state, list):
        return [_clone_state(s) for s in state]
    return state.clone()


def _copy_state_(dst: Any, src: Any) -> None:
    """Recursive in-place copy into STABLE state tensors (hip-graph
    replays read/write fixed addresses)."""
    if isinstance(dst, (tuple, list)):
        for d, s in zip(dst, src):
            _copy_state_(d, s)
    else:
        dst.copy_(src)


class RecPPOLearner:
    is_recurrent = True

    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs

        obs_space, act_space = env.observation_space, env.action_space
        self.actor = build_recurrent_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_recurrent_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        capturable = device.type == "cuda"
        self.actor_opt = torch.optim.Adam(
            self.actor.parameters(), lr=float(self.sys.actor_lr), eps=1e-5, capturable=capturable
        )
        self.critic_opt = torch.optim.Adam(
            self.critic.parameters(), lr=float(self.sys.critic_lr), eps=1e-5, capturable=capturable
        )
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()), device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 601)
        from stoix_amd.utils.training import maybe_lr_decay

        self.lr_decay = maybe_lr_decay(config, self.actor_opt, self.critic_opt)

        self.ts = env.reset()
        # STABLE hidden-state / carry buffers (hip-graph capturable: the
        # rollout graph reads them at t=0 and copies the final values back
        # at the end, so replays chain like PPOLearner.cur_obs)
        self.a_state = self.actor.initial_state(self.B, device)
        self.c_state = self.critic.initial_state(self.B, device)
        self._init_a = _clone_state(self.a_state)
        self._init_c = _clone_state(self.c_state)
        self.prev_done = torch.zeros(self.B, dtype=torch.bool, device=device)
        self.cur_obs = self.ts.observation.clone()
        self.episode_metrics: Dict[str, Tensor] = {}
        self.collect_metrics = True

        obs_shape = obs_space.shape
        self._discrete = not hasattr(act_space, "shape") or len(getattr(act_space, "shape", ())) == 0
        act_shape = () if self._discrete else act_space.shape
        act_dtype = torch.long if self._discrete else torch.float32
        T, B = self.T, self.B
        z = lambda *s, dtype=torch.float32: torch.zeros(*s, dtype=dtype, device=device)
        self.buf_obs = z(T, B, *obs_shape)
        self.buf_action = z(T, B, *act_shape, dtype=act_dtype)
        self.buf_log_prob = z(T, B)
        self.buf_value = z(T, B)
        self.buf_reward = z(T, B)
        self.buf_discount = z(T, B)
        self.buf_truncated = z(T, B, dtype=torch.bool)
        self.buf_reset = z(T, B, dtype=torch.bool)
        self.buf_adv = z(T, B)
        self.buf_targets = z(T, B)
        self.perm_buf = torch.arange(B, device=device)

    # --------------------------------------------------------------- acting

    @property
    def act_fn(self):
        from stoix_amd.evaluator import make_recurrent_act_fn

        return make_recurrent_act_fn(self.actor, self.gen)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self.rollout_phase()
        for _ in range(int(self.sys.epochs)):
            self._new_perm()
            metrics = self.epoch_phase()
        if self.lr_decay is not None:
            self.lr_decay.step()
        return metrics

    @torch.no_grad()
    def rollout_phase(self) -> None:
        """Rollout + GAE into stable buffers (hip-graph capturable: reads
        cur_obs/prev_done/hidden states, writes them back at the end)."""
        T, B = self.T, self.B
        _copy_state_(self._init_a, self.a_state)
        _copy_state_(self._init_c, self.c_state)
        a_state, c_state = self.a_state, self.c_state
        prev_done = self.prev_done
        obs = self.cur_obs
        for t in range(T):
            resets = prev_done
            dist, a_state = self.actor(obs.unsqueeze(0), resets.unsqueeze(0), a_state)
            value, c_state = self.critic(obs.unsqueeze(0), resets.unsqueeze(0), c_state)
            action = dist.sample(self.gen).squeeze(0)
            logp = dist.log_prob(action.unsqueeze(0)).squeeze(0)
            next_ts = self.env.step(action)
            self.buf_obs[t] = obs
            self.buf_action[t] = action
            self.buf_log_prob[t] = logp
            self.buf_value[t] = value.squeeze(0)
            self.buf_reward[t] = next_ts.reward
            self.buf_discount[t] = next_ts.discount
            self.buf_truncated[t] = next_ts.truncated()
            self.buf_reset[t] = resets
            prev_done = next_ts.last()
            obs = next_ts.observation
            last_ts = next_ts
        # bootstrap value of the final observation with the final state
        last_val, _ = self.critic(obs.unsqueeze(0), prev_done.unsqueeze(0), c_state)
        last_val = last_val.squeeze(0)
        # write carries back into the stable buffers so replays chain
        self.cur_obs.copy_(obs)
        self.prev_done.copy_(prev_done)
        _copy_state_(self.a_state, a_state)
        _copy_state_(self.c_state, c_state)
        self.ts = last_ts
        if self.collect_metrics:
            em = last_ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

        d_t = self.buf_discount * float(self.sys.gamma)
        v_next = torch.cat([self.buf_value[1:], last_val.unsqueeze(0)], dim=0)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            self.buf_reward, d_t, float(self.sys.gae_lambda), self.buf_value,
            v_next, truncation_t=self.buf_truncated,
            standardize_advantages=bool(self.sys.standardize_advantages),
        )
        self.buf_adv.copy_(adv)
        self.buf_targets.copy_(targets)

    def _new_perm(self) -> None:
        self.perm_buf.copy_(
            torch.randperm(self.B, device=self.device, generator=self.gen)
        )

    def epoch_phase(self) -> Dict[str, Tensor]:
        """One epoch of env-axis minibatches reading perm_buf (the
        permutation is refreshed eagerly between graph replays)."""
        B = self.B
        n_mb = int(self.sys.num_minibatches)
        envs_per_mb = B // n_mb
        metrics: Dict[str, Tensor] = {}
        for mb in range(n_mb):
            idx = self.perm_buf[mb * envs_per_mb : (mb + 1) * envs_per_mb]
            o = self.buf_obs[:, idx]
            rst = self.buf_reset[:, idx]
            a0 = _index_state(self._init_a, idx)
            c0 = _index_state(self._init_c, idx)
            dist, _ = self.actor(o, rst, a0)
            new_logp = dist.log_prob(self.buf_action[:, idx])
            entropy = dist.entropy().mean()
            a_loss = ppo_clip_loss(
                new_logp.reshape(-1), self.buf_log_prob[:, idx].reshape(-1),
                self.buf_adv[:, idx].reshape(-1), float(self.sys.clip_eps),
            )
            actor_loss = a_loss - float(self.sys.ent_coef) * entropy
            value, _ = self.critic(o, rst, c0)
            v_loss = clipped_value_loss(
                value.reshape(-1), self.buf_value[:, idx].reshape(-1),
                self.buf_targets[:, idx].reshape(-1), float(self.sys.clip_eps),
            )
            critic_loss = float(self.sys.vf_coef) * v_loss

            self.actor_opt.zero_grad(set_to_none=True)
            self.critic_opt.zero_grad(set_to_none=True)
            (actor_loss + critic_loss).backward()
            self.reducer.reduce()
            self.reducer.wait()
            nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
            nn.utils.clip_grad_norm_(self.critic.parameters(), float(self.sys.max_grad_norm))
            self.actor_opt.step()
            self.critic_opt.step()
            metrics = {
                "actor_loss": a_loss.detach(),
                "value_loss": v_loss.detach(),
                "entropy": entropy.detach(),
            }
        return metrics

    # ------------------------------------------------------- graph support

    def prepare_for_graph_capture(self) -> None:
        self.gen = None
        self.collect_metrics = False
        self.reducer._stream = None
        if hasattr(self.env, "prepare_for_graph_capture"):
            self.env.prepare_for_graph_capture()

    def after_graph_replay(self) -> None:
        from stoix_amd.envs.env import latched_episode_metrics

        self.episode_metrics = latched_episode_metrics(self.env, self)

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "critic": dict(self.critic.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> RecPPOLearner:
    return RecPPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_rec_ppo.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
