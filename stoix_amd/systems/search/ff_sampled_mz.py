"""Anakin Sampled MuZero — continuous actions with a learned world model
(parity: /root/reference/stoix/systems/search/ff_sampled_mz.py).

ff_mz with: a continuous-action world model (dynamics consume the raw action
vector, tanh-normal policy head), sampled MCTS over K candidate actions per
node, and a sampled policy loss — the weighted log-likelihood of the root's
candidate actions under the model's policy towards the search visit weights.
Value/reward remain two-hot categorical with K-step unroll and 0.5 gradient
scaling through the dynamics.
"""
from __future__ import annotations

import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.buffers import TrajectoryBuffer
from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.model_based import ContinuousRewardBasedWorldModel
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.search.mcts import sampled_mcts_search
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.search.ff_mz import scale_gradient

Tensor = torch.Tensor

_HALF_LOG_2PI = 0.9189385332046727


def _normal_logp(a: Tensor, loc: Tensor, scale: Tensor) -> Tensor:
    """Summed diagonal-normal log-density (the model's policy is an
    unsquashed normal over the env's action space)."""
    z = (a - loc) / scale
    return (-0.5 * z * z - scale.log() - _HALF_LOG_2PI).sum(-1)


class SampledMZLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)
        self.unroll = int(self.sys.unroll_steps)
        self.n_step = int(self.sys.n_step)
        self.K = int(getattr(self.sys, "num_sampled_actions", 8))

        obs_dim = 1
        for s in env.observation_space.shape:
            obs_dim *= s
        self.act_dim = env.action_space.shape[0]
        net_cfg = dict(getattr(config.network, "world_model", {}) or {})
        self.model = ContinuousRewardBasedWorldModel(obs_dim, self.act_dim, **net_cfg).to(device)
        broadcast_module(self.model)
        self.opt = torch.optim.Adam(self.model.parameters(), lr=float(self.sys.lr))
        self.reducer = FlatGradReducer(self.model.parameters(), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 991)

        seq = self.unroll + self.n_step + 1
        self.buffer = TrajectoryBuffer(
            add_batch_size=self.B,
            max_length_time_axis=max(seq + 1, int(self.sys.buffer_size) // self.B),
            sample_sequence_length=seq,
            device=device,
            seed=int(config.arch.seed) + 59,
        )
        self.batch_size = int(self.sys.batch_size)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    # ------------------------------------------------------------- search

    def _sample_from(self, loc: Tensor, scale: Tensor) -> Tensor:
        eps = torch.randn((loc.shape[0], self.K, self.act_dim), device=loc.device, generator=self.gen)
        return loc.unsqueeze(1) + scale.unsqueeze(1) * eps

    def _recurrent_fn(self, embedding: Dict[str, Tensor], action: Tensor):
        out = self.model.recurrent_inference([embedding["h"]], action)
        discount = torch.full_like(out.reward, self.gamma)
        cand = self._sample_from(out.policy_loc, out.policy_scale)
        return {"h": out.rnn_state[0]}, out.reward, discount, cand, out.value

    @torch.no_grad()
    def _search(self, obs: Tensor, greedy: bool = False):
        init = self.model.initial_inference(obs)
        cand = self._sample_from(init.policy_loc, init.policy_scale)
        return sampled_mcts_search(
            obs,
            {"h": init.rnn_state[0]},
            cand,
            init.value,
            self._recurrent_fn,
            num_simulations=int(self.sys.num_simulations),
            c_puct=float(getattr(self.sys, "c_puct", 1.25)),
            temperature=0.0 if greedy else float(getattr(self.sys, "search_temperature", 1.0)),
            generator=self.gen,
        )

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        return self._search(obs, greedy=True).action

    # -------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout_into_buffer(self) -> None:
        ts = self.ts
        obs_l, act_l, cand_l, w_l, sv_l, rew_l, disc_l = [], [], [], [], [], [], []
        for _ in range(self.T):
            obs = ts.observation
            out = self._search(obs)
            next_ts = self.env.step(out.action)
            obs_l.append(obs.clone())
            act_l.append(out.action)
            cand_l.append(out.sampled_actions)
            w_l.append(out.action_weights)
            sv_l.append(out.search_value)
            rew_l.append(next_ts.reward)
            disc_l.append(next_ts.discount)
            ts = next_ts
        self.ts = ts
        self.buffer.add(
            {
                "obs": torch.stack(obs_l, 1),
                "action": torch.stack(act_l, 1),
                "cand": torch.stack(cand_l, 1),
                "search_policy": torch.stack(w_l, 1),
                "search_value": torch.stack(sv_l, 1),
                "reward": torch.stack(rew_l, 1),
                "discount": torch.stack(disc_l, 1),
            }
        )
        em = ts.extras["episode_metrics"]
        final, has = get_final_step_metrics(em)
        if has:
            self.episode_metrics = {k: v.mean() for k, v in final.items()}

    # ---------------------------------------------------------------- update

    def _sampled_policy_loss(self, loc: Tensor, scale: Tensor, cand: Tensor, w: Tensor) -> Tensor:
        """-sum_k w_k log p(a_k | loc, scale), per sample. cand [B,K,adim]."""
        logp = _normal_logp(cand, loc.unsqueeze(1), scale.unsqueeze(1))  # [B, K]
        return -(w * logp).sum(-1)

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer()
        if not self.buffer.can_sample:
            return {}
        metrics: Dict[str, Tensor] = {}
        K, n = self.unroll, self.n_step
        for _ in range(int(self.sys.epochs)):
            batch = self.buffer.sample(self.batch_size)
            with torch.no_grad():
                r = batch["reward"].transpose(0, 1)
                d = batch["discount"].transpose(0, 1) * self.gamma
                sv = batch["search_value"].transpose(0, 1)
                z = multistep.batch_n_step_bootstrapped_returns(r, d, sv, n).transpose(0, 1)
                done = batch["discount"] == 0.0
                valid = torch.cumprod(1.0 - done.float(), dim=1)
                valid = torch.cat([torch.ones_like(valid[:, :1]), valid[:, :-1]], dim=1)

            init = self.model.initial_inference(batch["obs"][:, 0])
            state = init.rnn_state
            total_loss = torch.zeros((), device=self.device)
            pol_loss_acc = torch.zeros((), device=self.device)
            val_loss_acc = torch.zeros((), device=self.device)
            rew_loss_acc = torch.zeros((), device=self.device)
            pol0 = self._sampled_policy_loss(
                init.policy_loc, init.policy_scale, batch["cand"][:, 0], batch["search_policy"][:, 0]
            )
            val0 = self.model.value_head.ce_loss(init.value_logits, z[:, 0])
            total_loss = total_loss + (pol0 + val0).mean()
            pol_loss_acc = pol_loss_acc + pol0.mean()
            val_loss_acc = val_loss_acc + val0.mean()
            for k in range(1, K + 1):
                out = self.model.recurrent_inference(state, batch["action"][:, k - 1])
                state = [scale_gradient(s, 0.5) for s in out.rnn_state]
                m = valid[:, k]
                pol = self._sampled_policy_loss(
                    out.policy_loc, out.policy_scale, batch["cand"][:, k], batch["search_policy"][:, k]
                )
                val = self.model.value_head.ce_loss(out.value_logits, z[:, k])
                rew = self.model.reward_head.ce_loss(out.reward_logits, batch["reward"][:, k - 1])
                total_loss = total_loss + ((pol + val + rew) * m).mean()
                pol_loss_acc = pol_loss_acc + (pol * m).mean()
                val_loss_acc = val_loss_acc + (val * m).mean()
                rew_loss_acc = rew_loss_acc + (rew * m).mean()
            total_loss = total_loss / (K + 1)

            self.opt.zero_grad(set_to_none=True)
            total_loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.model.parameters(), float(self.sys.max_grad_norm))
            self.opt.step()
            metrics = {
                "total_loss": total_loss.detach(),
                "policy_loss": (pol_loss_acc / (K + 1)).detach(),
                "value_loss": (val_loss_acc / (K + 1)).detach(),
                "reward_loss": (rew_loss_acc / K).detach(),
            }
        return metrics

    def state_for_checkpoint(self):
        return {"model": dict(self.model.state_dict())}

    def snapshot_params(self):
        return {"model": {k: v.clone() for k, v in self.model.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.model.load_state_dict(snap["model"])


def learner_factory(config, env, device) -> SampledMZLearner:
    return SampledMZLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose(
        "default/anakin/default_ff_sampled_mz.yaml", argv if argv is not None else sys.argv[1:]
    )
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
