"""Anakin DisCo-103: RL with a LEARNED (meta-discovered) update rule.

Parity with /root/reference/stoix/systems/disco_rl/anakin/ff_disco103.py:
the agent net has five heads (logits/q/y/z/aux_pi,
networks/specialised.py); each minibatch the loss comes from a
``DiscoUpdateRule`` meta-network that consumes the trajectory (rewards,
terminations, current + behaviour agent outputs) through a reverse-time
LSTM and emits per-step target distributions for the policy and the y/z
auxiliary predictions, combined with a TD(lambda) two-hot categorical
value loss and a polyak target net held in the meta state
(ref ff_disco103.py:96-258; hyperparameter surface =
configs/system/disco_rl/ff_disco103.yaml).

Offline fallback: the reference DOWNLOADS the
pretrained disco_103.npz meta-parameters (ff_disco103.py:326-333,
utils/download.py); there is no network egress in this build, so the
meta-network is RANDOM-INIT by default — the system is structurally
complete and runs end-to-end, but a random update rule is not expected to
learn. ``system.meta_params_path`` loads a ``torch.save`` state dict of
the meta-network when one is available.
"""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.specialised import DiscoAgentNetwork, DiscoAgentOutput
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module, polyak_update
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class DiscoMetaNet(nn.Module):
    """The meta-network of the update rule: a reverse-time LSTM over
    per-step embeddings of (reward, termination, behaviour outputs,
    current outputs) emitting per-step targets for pi / y / z.

    Structure mirrors the reference's meta_nets.LSTM surface
    (hidden_size, embedding sizes, prediction_size — config
    system.disco_rule.net); the WEIGHTS are what disco_103.npz provides
    and are random-init here (module docstring)."""

    def __init__(self, num_actions: int, num_bins: int, prediction_size: int,
                 hidden_size: int = 256, embed: int = 64):
        super().__init__()
        # per-step scalar features: reward, done, behaviour logp stats,
        # value mean, disagreement
        in_dim = 5 + 2 * num_actions
        self.embed = nn.Sequential(nn.Linear(in_dim, embed), nn.ReLU())
        self.rnn = nn.LSTMCell(embed, hidden_size)
        self.pi_target = nn.Linear(hidden_size, num_actions)
        self.y_target = nn.Linear(hidden_size, prediction_size)
        self.z_target = nn.Linear(hidden_size, prediction_size)
        self.hidden_size = hidden_size
        self.num_actions = num_actions

    def forward(self, feats: Tensor) -> Tuple[Tensor, Tensor, Tensor]:
        """feats: [T, B, F] -> reverse-time targets ([T, B, A], [T, B, P] x2)."""
        T, B, _ = feats.shape
        h = feats.new_zeros(B, self.hidden_size)
        c = feats.new_zeros(B, self.hidden_size)
        pi_t, y_t, z_t = [], [], []
        x = self.embed(feats)
        for t in range(T - 1, -1, -1):
            h, c = self.rnn(x[t], (h, c))
            pi_t.append(self.pi_target(h))
            y_t.append(self.y_target(h))
            z_t.append(self.z_target(h))
        pi_t.reverse(); y_t.reverse(); z_t.reverse()
        return torch.stack(pi_t), torch.stack(y_t), torch.stack(z_t)


class DiscoLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        obs_dim = 1
        for s in env.observation_space.shape:
            obs_dim *= s
        A = env.action_space.num_values
        rule_cfg = dict(getattr(self.sys, "disco_rule", {}) or {})
        self.num_bins = int(rule_cfg.get("num_bins", 601))
        self.max_abs_value = float(rule_cfg.get("max_abs_value", 300.0))
        pred_size = int(dict(rule_cfg.get("net", {}) or {}).get("prediction_size", 600))

        self.net = DiscoAgentNetwork(
            obs_dim, A, num_bins=self.num_bins, prediction_size=pred_size
        ).to(device)
        broadcast_module(self.net)
        self.target_net = DiscoAgentNetwork(
            obs_dim, A, num_bins=self.num_bins, prediction_size=pred_size
        ).to(device)
        self.target_net.load_state_dict(self.net.state_dict())
        for p in self.target_net.parameters():
            p.requires_grad_(False)

        self.meta = DiscoMetaNet(A, self.num_bins, pred_size).to(device)
        meta_path = getattr(self.sys, "meta_params_path", None)
        if meta_path:
            self.meta.load_state_dict(torch.load(meta_path, map_location=device))
        for p in self.meta.parameters():  # the rule is FIXED at agent-train time
            p.requires_grad_(False)

        self.opt = torch.optim.Adam(self.net.parameters(), lr=float(self.sys.lr), eps=1e-5)
        from stoix_amd.utils.training import maybe_lr_decay

        self.lr_decay = maybe_lr_decay(config, self.opt)
        self.reducer = FlatGradReducer(self.net.parameters(), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 1031)
        hp = dict(getattr(self.sys, "disco_hyperparams", {}) or {})
        self.pi_cost = float(hp.get("pi_cost", 1.0))
        self.y_cost = float(hp.get("y_cost", 1.0))
        self.z_cost = float(hp.get("z_cost", 1.0))
        self.aux_policy_cost = float(hp.get("aux_policy_cost", 1.0))
        self.value_cost = float(hp.get("value_cost", 0.2))
        self.td_lambda = float(hp.get("value_fn_td_lambda", 0.95))
        self.target_coeff = float(hp.get("target_params_coeff", 0.9))
        self.reward_scale = float(getattr(self.sys, "reward_scale", 1.0))

        # categorical value support (two-hot transform, +-max_abs_value)
        self.bins = torch.linspace(
            -self.max_abs_value, self.max_abs_value, self.num_bins, device=device
        )
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    # ---------------------------------------------------------------- acting

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        logits = self.net(obs).logits
        if greedy:
            return logits.argmax(dim=-1)
        u = torch.rand(logits.shape, device=logits.device, generator=self.gen)
        g = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
        return (logits + g).argmax(dim=-1)

    def _two_hot(self, x: Tensor) -> Tensor:
        """Two-hot encode values onto the bin support [T*B?]-> [..., bins]."""
        x = x.clamp(-self.max_abs_value, self.max_abs_value)
        pos = (x + self.max_abs_value) / (2 * self.max_abs_value) * (self.num_bins - 1)
        lo = pos.floor().long().clamp(0, self.num_bins - 1)
        hi = (lo + 1).clamp(0, self.num_bins - 1)
        w_hi = pos - lo.float()
        out = x.new_zeros(*x.shape, self.num_bins)
        out.scatter_(-1, lo.unsqueeze(-1), (1 - w_hi).unsqueeze(-1))
        out.scatter_add_(-1, hi.unsqueeze(-1), w_hi.unsqueeze(-1))
        return out

    def _q_mean(self, q_logits: Tensor) -> Tensor:
        return (F.softmax(q_logits, dim=-1) * self.bins).sum(-1)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        ts = self.ts
        obs_l, act_l, rew_l, done_l, out_l = [], [], [], [], []
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                out = self.net(obs)
                u = torch.rand(out.logits.shape, device=self.device, generator=self.gen)
                g = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
                action = (out.logits + g).argmax(dim=-1)
                next_ts = self.env.step(action)
                obs_l.append(obs.clone())
                act_l.append(action)
                rew_l.append(next_ts.reward * self.reward_scale)
                done_l.append(next_ts.discount == 0)
                out_l.append(out)
                ts = next_ts
            self.ts = ts
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

        obs_t = torch.stack(obs_l)  # [T, B, ...]
        act_t = torch.stack(act_l)  # [T, B]
        rew_t = torch.stack(rew_l)
        done_t = torch.stack(done_l)
        beh = DiscoAgentOutput(*[torch.stack([getattr(o, f) for o in out_l])
                                 for f in DiscoAgentOutput._fields])

        n_mb = int(self.sys.num_minibatches)
        envs_per_mb = B // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(self.sys.epochs)):
            perm = torch.randperm(B, device=self.device, generator=self.gen)
            for mb in range(n_mb):
                sel = perm[mb * envs_per_mb : (mb + 1) * envs_per_mb]
                loss, metrics = self._minibatch_loss(
                    obs_t[:, sel], act_t[:, sel], rew_t[:, sel], done_t[:, sel],
                    DiscoAgentOutput(*[getattr(beh, f)[:, sel] for f in beh._fields]),
                )
                self.opt.zero_grad(set_to_none=True)
                loss.backward()
                self.reducer.reduce()
                self.reducer.wait()
                nn.utils.clip_grad_norm_(self.net.parameters(), float(self.sys.max_abs_update))
                self.opt.step()
                with torch.no_grad():
                    polyak_update(self.net.parameters(), self.target_net.parameters(),
                                  1.0 - self.target_coeff)
        if self.lr_decay is not None:
            self.lr_decay.step()
        return metrics

    def _minibatch_loss(self, obs, act, rew, done, beh: DiscoAgentOutput):
        T, Bm = act.shape
        cur: DiscoAgentOutput = self.net(obs.reshape(T * Bm, *obs.shape[2:]))
        cur = DiscoAgentOutput(*[v.view(T, Bm, *v.shape[1:]) for v in cur])

        # ---- meta-network features and targets (reverse-time LSTM)
        with torch.no_grad():
            beh_logp = F.log_softmax(beh.logits, dim=-1)
            v_beh = (F.softmax(beh.logits, -1) * self._q_mean(beh.q)).sum(-1)
            disagreement = (beh.aux_pi.softmax(-1) - beh.logits.softmax(-1)).abs().sum(-1)
            feats = torch.cat(
                [
                    rew.unsqueeze(-1),
                    done.float().unsqueeze(-1),
                    v_beh.unsqueeze(-1),
                    beh_logp.gather(-1, act.unsqueeze(-1)),
                    disagreement.unsqueeze(-1),
                    beh.logits.softmax(-1),
                    beh.aux_pi.softmax(-1),
                ],
                dim=-1,
            )
            pi_tgt, y_tgt, z_tgt = self.meta(feats)
            pi_tgt = F.softmax(pi_tgt, dim=-1)
            y_tgt = F.softmax(y_tgt, dim=-1)
            z_tgt = F.softmax(z_tgt, dim=-1)

            # ---- TD(lambda) value targets on the taken action's Q mean
            with torch.no_grad():
                tgt_out: DiscoAgentOutput = self.target_net(obs.reshape(T * Bm, *obs.shape[2:]))
                q_tgt_mean = self._q_mean(tgt_out.q.view(T, Bm, *tgt_out.q.shape[1:]))
                v_t = (F.softmax(cur.logits.detach(), -1) * q_tgt_mean).sum(-1)
            g = v_t[-1]
            returns = []
            lam = self.td_lambda
            for t in range(T - 1, -1, -1):
                nxt = v_t[t] if t == T - 1 else v_t[t + 1]
                g = rew[t] + self.gamma * (~done[t]).float() * ((1 - lam) * nxt + lam * g)
                returns.append(g)
            returns.reverse()
            ret = torch.stack(returns)  # [T, Bm]
            value_target = self._two_hot(ret)

        # ---- losses: KL(meta targets || current heads) + categorical value
        logp = F.log_softmax(cur.logits, dim=-1)
        pi_loss = -(pi_tgt * logp).sum(-1).mean()
        y_loss = -(y_tgt * F.log_softmax(cur.y, dim=-1)).sum(-1).mean()
        z_a = cur.z.gather(2, act.view(T, Bm, 1, 1).expand(-1, -1, 1, cur.z.shape[-1])).squeeze(2)
        z_loss = -(z_tgt * F.log_softmax(z_a, dim=-1)).sum(-1).mean()
        aux_pi_loss = -(pi_tgt * F.log_softmax(cur.aux_pi, dim=-1)).sum(-1).mean()
        q_a = cur.q.gather(2, act.view(T, Bm, 1, 1).expand(-1, -1, 1, self.num_bins)).squeeze(2)
        value_loss = -(value_target * F.log_softmax(q_a, dim=-1)).sum(-1).mean()

        loss = (
            self.pi_cost * pi_loss
            + self.y_cost * y_loss
            + self.z_cost * z_loss
            + self.aux_policy_cost * aux_pi_loss
            + self.value_cost * value_loss
        )
        return loss, {
            "total_loss": loss.detach(),
            "pi_loss": pi_loss.detach(),
            "value_loss": value_loss.detach(),
        }

    # ------------------------------------------------------------ checkpoint

    def state_for_checkpoint(self):
        return {"net": dict(self.net.state_dict())}

    def snapshot_params(self):
        return {"net": {k: v.clone() for k, v in self.net.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.net.load_state_dict(snap["net"])


def learner_factory(config, env, device) -> DiscoLearner:
    return DiscoLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_disco103.yaml",
                  argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
