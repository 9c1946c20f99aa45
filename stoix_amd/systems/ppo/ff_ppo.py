"""Anakin PPO (feed-forward) — discrete and continuous.

Algorithm parity with /root/reference/stoix/systems/ppo/anakin/ff_ppo.py and
ff_ppo_continuous.py (the head is chosen by the network config): rollout of
``rollout_length`` steps storing PPOTransition with ``bootstrap_value =
V(extras["next_obs"])`` (ff_ppo.py:113-116), truncation-aware GAE
(ff_ppo.py:164-179), epochs x shuffled minibatches of clip-loss + entropy
and clipped value loss (ff_ppo.py:191-235, 296-336), separate Adam chains
with global-norm clip (ff_ppo.py:456-463), optional Welford observation
normalisation (ff_ppo.py:90-162).

MI355X design: rollout storage is preallocated [T, B, ...] device tensors
(struct-of-arrays); the rollout and update loops are static-shaped and
capturable into hip graphs (stoix_amd/ops/graph.py); gradients are averaged
across GPUs with one fused flat RCCL all-reduce per minibatch
(stoix_amd.parallel.FlatGradReducer).
"""
from __future__ import annotations

import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.ops import running_statistics as rs
from stoix_amd.ops.losses import clipped_value_loss, ppo_clip_loss
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class PPOLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs

        obs_space, act_space = env.observation_space, env.action_space
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
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
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 13)

        # bf16 compute path (BASELINE config "Brax Ant bf16"): network fwd/bwd
        # run under autocast-bf16 (MFMA-rate GEMMs); env physics, GAE, losses
        # accumulate in fp32; optimiser state fp32.
        use_bf16 = str(getattr(self.sys, "compute_dtype", "fp32")) == "bf16"
        self.amp = (
            torch.autocast(device_type="cuda", dtype=torch.bfloat16)
            if use_bf16 and device.type == "cuda"
            else None
        )

        self.normalize_obs = bool(getattr(self.sys, "normalize_observations", False))
        if self.normalize_obs:
            self.obs_stats = rs.init_state(obs_space.shape, device=device)

        self.ts = env.reset()
        # stable-address current-observation buffer: the graph-captured
        # rollout reads it at t=0 and writes the final obs back at the end,
        # so replays chain correctly (ops/graph.py).
        self.cur_obs = self.ts.observation.clone()
        self.collect_metrics = True
        self.episode_metrics: Dict[str, float] = {}
        self._discrete = not hasattr(act_space, "shape") or len(act_space.shape) == 0

        # preallocated rollout storage (struct of arrays, time-major)
        obs_shape = obs_space.shape
        act_shape = () if self._discrete else act_space.shape
        act_dtype = torch.long if self._discrete else torch.float32
        z = lambda *s, dtype=torch.float32: torch.zeros(*s, dtype=dtype, device=device)
        self.buf_obs = z(self.T, self.B, *obs_shape)
        self.buf_action = z(self.T, self.B, *act_shape, dtype=act_dtype)
        self.buf_log_prob = z(self.T, self.B)
        self.buf_value = z(self.T, self.B)
        self.buf_bootstrap = z(self.T, self.B)
        self.buf_reward = z(self.T, self.B)
        self.buf_discount = z(self.T, self.B)
        self.buf_truncated = z(self.T, self.B, dtype=torch.bool)
        self.buf_adv = z(self.T, self.B)
        self.buf_targets = z(self.T, self.B)
        self.perm_buf = torch.arange(self.T * self.B, device=device)

        # linear LR decay by update count (reference utils/training.py:6-49)
        from stoix_amd.utils.training import maybe_lr_decay

        self.lr_decay = maybe_lr_decay(config, self.actor_opt, self.critic_opt)

        # fused MI355X path (hand-written MFMA/HIP kernels; see fused.py):
        # replaces the eager rollout + minibatch update when the network is
        # the canonical MLP shape. system.fused=false forces eager.
        self.fused = None
        if bool(getattr(self.sys, "fused", True)) and use_bf16 and self.lr_decay is None:
            from stoix_amd.systems.ppo.fused import FusedPPOEngine

            self.fused = FusedPPOEngine.try_build(self)

    # ---------------------------------------------------------------- acting

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        if self.normalize_obs:
            obs = rs.normalize(obs, self.obs_stats)
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # --------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout(self) -> None:
        obs_raw = self.cur_obs
        for t in range(self.T):
            obs = obs_raw  # buf stores RAW obs (reference ff_ppo.py:148-162:
            # the trajectory keeps raw observations; normalisation for the
            # update happens post-rollout with PRE-update statistics)
            if self.normalize_obs:
                obs = rs.normalize(obs, self.obs_stats)
            if self.amp is not None:
                with self.amp:
                    dist = self.actor(obs)
                    value = self.critic(obs)
                    action = dist.sample(self.gen)
                    log_prob = dist.log_prob(action)
            else:
                dist = self.actor(obs)
                value = self.critic(obs)
                action = dist.sample(self.gen)
                log_prob = dist.log_prob(action)
            next_ts = self.env.step(action)
            next_obs = next_ts.extras["next_obs"]
            if self.normalize_obs:
                next_obs = rs.normalize(next_obs, self.obs_stats)
            if self.amp is not None:
                with self.amp:
                    bootstrap_value = self.critic(next_obs)
            else:
                bootstrap_value = self.critic(next_obs)

            self.buf_obs[t] = obs_raw
            self.buf_action[t] = action
            self.buf_log_prob[t] = log_prob.float()
            self.buf_value[t] = value.float()
            self.buf_bootstrap[t] = bootstrap_value.float()
            self.buf_reward[t] = next_ts.reward
            self.buf_discount[t] = next_ts.discount
            self.buf_truncated[t] = next_ts.truncated()
            obs_raw = next_ts.observation
            last_ts = next_ts
        self.cur_obs.copy_(obs_raw)
        if self.normalize_obs:
            # normalise the stored trajectory with the PRE-update statistics,
            # THEN update the statistics from the raw observations
            # (reference ff_ppo.py:150-162 ordering)
            normalized = rs.normalize(self.buf_obs, self.obs_stats)
            self.obs_stats = rs.update(self.obs_stats, self.buf_obs, all_reduce=True)
            self.buf_obs.copy_(normalized)
        if self.collect_metrics:
            em = last_ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

    # ---------------------------------------------------------------- losses

    def policy_loss(self, new_logp: Tensor, old_logp: Tensor, adv: Tensor) -> Tensor:
        """PPO clipped surrogate; subclasses override (penalty / drift)."""
        return ppo_clip_loss(new_logp, old_logp, adv, float(self.sys.clip_eps))

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self.rollout_phase()
        for _ in range(int(self.sys.epochs)):
            self._new_perm()
            metrics = self.epoch_phase()
        if self.lr_decay is not None:
            self.lr_decay.step()
        return metrics

    def rollout_phase(self) -> None:
        """Rollout + GAE into stable buffers (hip-graph capturable)."""
        if self.fused is not None:
            self.fused.rollout()
        else:
            self._rollout()
        gamma = float(self.sys.gamma)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            self.buf_reward,
            gamma * self.buf_discount,
            float(self.sys.gae_lambda),
            self.buf_value,
            self.buf_bootstrap,
            truncation_t=self.buf_truncated,
            standardize_advantages=bool(self.sys.standardize_advantages),
        )
        self.buf_adv.copy_(adv)
        self.buf_targets.copy_(targets)

    def _new_perm(self) -> None:
        TB = self.T * self.B
        self.perm_buf.copy_(torch.randperm(TB, device=self.device, generator=self.gen))

    def epoch_phase(self) -> Dict[str, Tensor]:
        """One epoch of minibatch updates reading self.perm_buf
        (hip-graph capturable; the permutation is refreshed eagerly between
        replays — randperm is not capture-legal at scale)."""
        if self.fused is not None:
            return self.fused.epoch()
        TB = self.T * self.B
        flat_obs = self.buf_obs.reshape(TB, *self.buf_obs.shape[2:])
        flat_action = self.buf_action.reshape(TB, *self.buf_action.shape[2:])
        flat_logp = self.buf_log_prob.reshape(TB)
        flat_value = self.buf_value.reshape(TB)
        flat_adv = self.buf_adv.reshape(TB)
        flat_targets = self.buf_targets.reshape(TB)

        n_mb = int(self.sys.num_minibatches)
        mb_size = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        if True:
            perm = self.perm_buf
            for mb in range(n_mb):
                idx = perm[mb * mb_size : (mb + 1) * mb_size]
                obs_mb = flat_obs[idx]
                if self.amp is not None:
                    with self.amp:
                        dist = self.actor(obs_mb)
                        new_logp = dist.log_prob(flat_action[idx])
                        entropy = dist.entropy().mean()
                        value = self.critic(obs_mb)
                    new_logp = new_logp.float()
                    entropy = entropy.float()
                    value = value.float()
                else:
                    dist = self.actor(obs_mb)
                    new_logp = dist.log_prob(flat_action[idx])
                    entropy = dist.entropy().mean()
                    value = self.critic(obs_mb)
                a_loss = self.policy_loss(new_logp, flat_logp[idx], flat_adv[idx])
                actor_loss = a_loss - float(self.sys.ent_coef) * entropy

                v_loss = clipped_value_loss(value, flat_value[idx], flat_targets[idx], float(self.sys.clip_eps))
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
        """Switch to capture-safe modes: default CUDA RNG (graph-aware),
        inline all-reduce on the capture stream, no host-side metric
        reads; capture-safe torch envs switch to unconditional-autoreset
        graph mode."""
        self.gen = None
        self.collect_metrics = False
        self.reducer._stream = None
        if hasattr(self.env, "prepare_for_graph_capture"):
            self.env.prepare_for_graph_capture()

    def after_graph_replay(self) -> None:
        # episode metrics read eagerly from the env's latched buffers, with
        # a device freshness flag (resolved at log time)
        from stoix_amd.envs.env import latched_episode_metrics

        self.episode_metrics = latched_episode_metrics(self.env, self)

    # ------------------------------------------------------------ checkpoint

    def state_for_checkpoint(self):
        return {
            "actor": {k: v for k, v in self.actor.state_dict().items()},
            "critic": {k: v for k, v in self.critic.state_dict().items()},
        }

    def snapshot_params(self):
        return {
            "actor": {k: v.clone() for k, v in self.actor.state_dict().items()},
            "critic": {k: v.clone() for k, v in self.critic.state_dict().items()},
        }

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])
        self.critic.load_state_dict(snap["critic"])
        if self.fused is not None:
            self.fused.refresh_masters()

    def aux_checkpoint_state(self) -> dict:
        """Optimizer moments for true resume (the reference's LearnerState
        checkpoints include opt_states; template restore can't rebuild a
        fresh optimizer's empty slots, so these ride as the aux payload)."""
        return {
            "actor_opt": self.actor_opt.state_dict(),
            "critic_opt": self.critic_opt.state_dict(),
        }

    def load_aux_checkpoint_state(self, aux: dict) -> None:
        self.actor_opt.load_state_dict(aux["actor_opt"])
        self.critic_opt.load_state_dict(aux["critic_opt"])


def learner_factory(config, env, device) -> PPOLearner:
    return PPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    overrides = argv if argv is not None else sys.argv[1:]
    cfg = compose("default/anakin/default_ff_ppo.yaml", overrides)
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
