"""Anakin Rainbow (parity: /root/reference/stoix/systems/q_learning/
ff_rainbow.py): noisy dueling distributional net with fresh noise per apply
(:176-186), prioritised trajectory buffer sampling n-step windows
(:433-444), n-step rewards via discounted returns on the window (:231-246),
IS weights (1/p)^beta / max with beta annealed to 1 (:381-390, 202-205),
new priorities = per-sample categorical TD error written back (:262-266).
"""
from __future__ import annotations

import copy
import sys
from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.buffers import PrioritisedBuffer
from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.dueling import DistributionalDuelingQNetwork
from stoix_amd.networks.layers import NoiseBank, set_noise_enabled
from stoix_amd.ops.losses import categorical_l2_project
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class RainbowLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)
        self.n_step = int(self.sys.n_step)

        obs_dim = 1
        for s in env.observation_space.shape:
            obs_dim *= s
        net_cfg = dict(config.network.actor_network.get("action_head", {}))
        net_cfg.pop("_target_", None)
        self.q_online = DistributionalDuelingQNetwork(
            obs_dim, env.action_space.num_values, noisy=True, **net_cfg
        ).to(device)
        broadcast_module(self.q_online)
        self.q_target = copy.deepcopy(self.q_online)
        for p in self.q_target.parameters():
            p.requires_grad_(False)
        # fused noise resampling: 3 kernels per net instead of 4 x 2 x layers
        self.noise_online = NoiseBank(self.q_online)
        self.noise_target = NoiseBank(self.q_target)
        # bf16 compute path: GEMMs at MFMA rate under autocast; the C51
        # projection/CE and the priorities stay fp32
        use_bf16 = str(getattr(self.sys, "compute_dtype", "fp32")) == "bf16"
        self._use_amp = use_bf16 and device.type == "cuda"

        self.opt = torch.optim.Adam(
            self.q_online.parameters(), lr=float(self.sys.q_lr), eps=1e-5,
            capturable=device.type == "cuda",
        )
        self.reducer = FlatGradReducer(self.q_online.parameters(), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 503)

        self.buffer = PrioritisedBuffer(
            add_batch_size=self.B,
            max_length_time_axis=max(self.n_step + 1, int(self.sys.buffer_size) // self.B),
            sample_sequence_length=self.n_step,
            device=device,
            seed=int(config.arch.seed) + 43,
            priority_exponent=float(self.sys.priority_exponent),
        )
        self.batch_size = int(self.sys.batch_size)
        self.beta0 = float(self.sys.importance_sampling_exponent)
        self.total_updates = max(1, int(config.arch.num_updates) * int(self.sys.epochs))
        # device update counter: the beta anneal is computed on-device so
        # the whole update step stays hip-graph capturable
        self._update_count = torch.zeros((), device=device)

        self.ts = env.reset()
        self.collect_metrics = True
        self.episode_metrics: Dict[str, Tensor] = {}
        self._warmup()

    # --------------------------------------------------------------- acting

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        # eval uses the deterministic net (no noise), greedy argmax
        set_noise_enabled(self.q_online, False)
        q = self.q_online(obs).q_values
        set_noise_enabled(self.q_online, True)
        return q.argmax(dim=-1)

    @torch.no_grad()
    def _rollout_into_buffer(self, steps: int, random_actions: bool = False) -> None:
        ts = self.ts
        obs_l, act_l, rew_l, disc_l, next_l = [], [], [], [], []
        for _ in range(steps):
            obs = ts.observation
            if random_actions:
                action = self.env.action_space.sample(self.B, self.device, self.gen)
            else:
                self.noise_online.resample(self.gen)  # noisy-net exploration
                if self._use_amp:
                    with torch.autocast("cuda", torch.bfloat16):
                        q = self.q_online(obs).q_values
                else:
                    q = self.q_online(obs).q_values
                action = q.argmax(dim=-1)
            next_ts = self.env.step(action)
            obs_l.append(obs.clone())
            act_l.append(action)
            rew_l.append(next_ts.reward.clamp(-float(self.sys.max_abs_reward), float(self.sys.max_abs_reward)))
            disc_l.append(next_ts.discount)
            next_l.append(next_ts.extras["next_obs"].clone())
            ts = next_ts
        self.ts = ts
        self.buffer.add(
            {
                "obs": torch.stack(obs_l, 1),
                "action": torch.stack(act_l, 1),
                "reward": torch.stack(rew_l, 1),
                "discount": torch.stack(disc_l, 1),
                "next_obs": torch.stack(next_l, 1),
            }
        )
        if self.collect_metrics:
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

    def _warmup(self) -> None:
        steps = max(self.n_step + 1, int(getattr(self.sys, "warmup_steps", 64)) // self.B + 1)
        self._rollout_into_buffer(steps, random_actions=True)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer(self.T)
        metrics: Dict[str, Tensor] = {}
        tau = float(self.sys.tau)
        for _ in range(int(self.sys.epochs)):
            beta = self.beta0 + (1.0 - self.beta0) * (
                self._update_count / self.total_updates
            ).clamp(max=1.0)
            batch = self.buffer.sample(self.batch_size, importance_sampling_exponent=beta)
            # n-step reward over the window; bootstrap from the window end
            r = batch["reward"]  # [B, n]
            d = batch["discount"] * self.gamma
            n_step_disc = torch.cumprod(d, dim=1)
            disc_prefix = torch.cat([torch.ones_like(d[:, :1]), n_step_disc[:, :-1]], dim=1)
            n_step_reward = (r * disc_prefix).sum(dim=1)
            bootstrap_disc = n_step_disc[:, -1]
            obs0 = batch["obs"][:, 0]
            act0 = batch["action"][:, 0]
            obs_n = batch["next_obs"][:, -1]

            self.noise_online.resample(self.gen)
            self.noise_target.resample(self.gen)
            # ONE online forward over [obs0; obs_n] (the double-Q selector
            # shares weights and noise with the TD forward; batching the
            # rows halves the online-net kernel count)
            obs_cat = torch.cat([obs0, obs_n], dim=0)
            Bn = obs0.shape[0]
            if self._use_amp:
                with torch.autocast("cuda", torch.bfloat16):
                    out_all = self.q_online(obs_cat)
                    with torch.no_grad():
                        out_t = self.q_target(obs_n)
            else:
                out_all = self.q_online(obs_cat)
                with torch.no_grad():
                    out_t = self.q_target(obs_n)
            out_tm1 = out_all
            with torch.no_grad():
                best_a = out_all.q_values[Bn:].argmax(dim=-1)
                probs_t = F.softmax(out_t.q_logits.float(), dim=-1)
                p_best = probs_t.gather(
                    1, best_a.view(-1, 1, 1).expand(-1, 1, probs_t.shape[-1])
                ).squeeze(1)
                target_z = n_step_reward.unsqueeze(-1) + bootstrap_disc.unsqueeze(-1) * out_t.atoms
                target = categorical_l2_project(target_z, p_best, out_tm1.atoms)
            logits_a = out_tm1.q_logits[:Bn].float().gather(
                1, act0.view(-1, 1, 1).expand(-1, 1, out_tm1.q_logits.shape[-1])
            ).squeeze(1)
            ce = -(target * F.log_softmax(logits_a, dim=-1)).sum(-1)  # [B]
            loss = (batch["_weights"] * ce).mean()

            self.opt.zero_grad(set_to_none=True)
            loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.q_online.parameters(), float(self.sys.max_grad_norm))
            self.opt.step()
            # priority writeback: per-sample TD error magnitude
            self.buffer.set_priorities(batch["_slots"], ce.detach())
            with torch.no_grad():
                tgt = list(self.q_target.parameters())
                torch._foreach_mul_(tgt, 1 - tau)
                torch._foreach_add_(tgt, list(self.q_online.parameters()), alpha=tau)
            self._update_count += 1
            metrics = {"q_loss": loss.detach(), "beta": beta.detach()}
        return metrics

    # ------------------------------------------------------- graph support

    @property
    def graph_capturable(self) -> bool:
        return (
            getattr(self.env, "_hip", None) is not None
            or getattr(self.env, "capture_safe", False)
        )

    def prepare_for_graph_capture(self) -> None:
        """Capture-safe modes: default (graph-aware) CUDA RNG for noisy-net
        resampling and buffer sampling, inline all-reduce, no host-side
        metric reads. The sum-tree update/sample and the beta anneal are
        already device-resident (buffers/per.py, ops/csrc/per.hip)."""
        self.gen = None
        self.collect_metrics = False
        self.reducer._stream = None
        self.buffer.graph_safe_rng = True

    def after_graph_replay(self) -> None:
        from stoix_amd.envs.env import latched_episode_metrics

        self.episode_metrics = latched_episode_metrics(self.env, self)

    def state_for_checkpoint(self):
        return {"q_online": dict(self.q_online.state_dict())}

    def snapshot_params(self):
        return {"q_online": {k: v.clone() for k, v in self.q_online.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.q_online.load_state_dict(snap["q_online"])


def learner_factory(config, env, device) -> RainbowLearner:
    return RainbowLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_rainbow.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
