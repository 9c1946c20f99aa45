"""Sebulba IMPALA (parity: /root/reference/stoix/systems/impala/sebulba/
ff_impala.py): actors store behaviour log-probs at act time; the learner
computes rho = exp(logpi_now - logpi_behaviour) and V-trace errors +
pg-advantages (vmapped rlax.vtrace_td_error_and_advantage, :426-440, here
the HIP vtrace kernel), losses = 0.5*sum(vtrace err^2) + pg + entropy
(:440-473); single pass minibatched over the env axis, grad-norm 40,
lr 6e-4 (ff_impala.yaml). Reuses the Sebulba PPO actor threads (the payload
already carries log_prob/value traces).
"""
from __future__ import annotations

import sys
import threading
import time
from typing import Dict, List

import torch
import torch.nn as nn

from stoix_amd import envs as environments
from stoix_amd.config import compose
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module, get_dist_context
from stoix_amd.systems.ppo.sebulba_ff_ppo import _device_of, actor_thread_fn
from stoix_amd.utils.logger import LogEvent, StoixLogger
from stoix_amd.utils.sebulba import (
    AsyncEvaluator,
    OnPolicyPipeline,
    ParameterServer,
    PinnedCopier,
    ThreadLifetime,
)
from stoix_amd.utils.timing import TimingTracker
from stoix_amd.utils.total_timestep_checker import check_total_timesteps

Tensor = torch.Tensor


class ImpalaLearner:
    def __init__(self, config, device: torch.device, obs_space, act_space):
        self.cfg = config
        self.sys = config.system
        self.device = device
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        params = list(self.actor.parameters()) + list(self.critic.parameters())
        self.opt = torch.optim.Adam(params, lr=float(self.sys.learner_lr), eps=1e-5)
        self.reducer = FlatGradReducer(params, device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 877)
        self.copier = PinnedCopier(device)

    def cpu_params(self) -> Dict[str, Dict[str, Tensor]]:
        return {
            "actor": {k: v.detach().cpu() for k, v in self.actor.state_dict().items()},
            "critic": {k: v.detach().cpu() for k, v in self.critic.state_dict().items()},
        }

    def learn(self, payloads: List[Dict[str, Tensor]]) -> Dict[str, Tensor]:
        sysc = self.sys
        merged: Dict[str, Tensor] = {}
        for k in ("obs", "action", "log_prob", "reward", "discount", "last_value"):
            merged[k] = torch.cat([p[k] for p in payloads], dim=-1 if payloads[0][k].dim() == 1 else 1)
        batch = self.copier.to_device(merged)
        T, B = batch["obs"].shape[:2]
        gamma = float(sysc.gamma)

        n_mb = int(sysc.num_minibatches)
        envs_per_mb = B // n_mb
        perm = torch.randperm(B, device=self.device, generator=self.gen)
        metrics: Dict[str, Tensor] = {}
        for i in range(n_mb):
            idx = perm[i * envs_per_mb : (i + 1) * envs_per_mb]
            obs = batch["obs"][:, idx]
            act = batch["action"][:, idx]
            b_logp = batch["log_prob"][:, idx]
            flat_obs = obs.reshape(T * envs_per_mb, *obs.shape[2:])
            dist = self.actor(flat_obs)
            logp = dist.log_prob(act.reshape(T * envs_per_mb, *act.shape[2:])).reshape(T, -1)
            entropy = dist.entropy().mean()
            v = self.critic(flat_obs).reshape(T, -1)
            with torch.no_grad():
                rho = torch.exp(logp.detach() - b_logp)
            v_next = torch.cat([v[1:].detach(), batch["last_value"][idx].unsqueeze(0)], dim=0)
            errors, pg_adv, _ = multistep.vtrace_td_error_and_advantage(
                v.detach(),
                v_next,
                batch["reward"][:, idx],
                gamma * batch["discount"][:, idx],
                rho,
                lambda_=float(sysc.vtrace_lambda),
                clip_rho_threshold=float(sysc.clip_rho_threshold),
                clip_pg_rho_threshold=float(sysc.clip_pg_rho_threshold),
            )
            vs = (errors + v.detach()).detach()
            critic_loss = 0.5 * ((v - vs) ** 2).sum(dim=0).mean()
            pg_loss = -(pg_adv.detach() * logp).sum(dim=0).mean()
            ent_loss = -entropy
            loss = pg_loss + float(sysc.vf_coef) * critic_loss + float(sysc.ent_coef) * ent_loss
            self.opt.zero_grad(set_to_none=True)
            loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            nn.utils.clip_grad_norm_(
                list(self.actor.parameters()) + list(self.critic.parameters()),
                float(sysc.max_grad_norm),
            )
            self.opt.step()
            metrics = {
                "pg_loss": pg_loss.detach(),
                "value_loss": critic_loss.detach(),
                "entropy": entropy.detach(),
            }
        return metrics


def run_experiment(config, force_cpu: bool = False) -> float:
    ctx = get_dist_context(force_cpu=force_cpu)
    config.arch.n_devices = ctx.world_size
    check_total_timesteps(config)
    torch.manual_seed(int(config.arch.seed) + ctx.rank)

    actor_device_ids = list(config.arch.actor.device_ids)
    per_dev = int(config.arch.actor.actor_per_device)
    n_actors = len(actor_device_ids) * per_dev
    learner_device = _device_of(config.arch.learner.device_ids[0], force_cpu)
    eval_device = _device_of(getattr(config.arch, "evaluator_device_id", 0), force_cpu)

    env_factory = environments.make_factory(config, device="cpu")
    probe_env = env_factory(1)
    obs_space, act_space = probe_env.observation_space, probe_env.action_space
    learner = ImpalaLearner(config, learner_device, obs_space, act_space)


    # restore-at-startup (reference sebulba ff_ppo.py:783-789)
    if getattr(config.logger.checkpointing, "load_model", False):
        from stoix_amd.utils.checkpointing import Checkpointer

        load_args = config.logger.checkpointing.load_args
        if not load_args.checkpoint_uid:
            raise ValueError(
                "load_model=true needs logger.checkpointing.load_args.checkpoint_uid"
            )
        loader = Checkpointer(
            model_name=config.system.system_name,
            directory=str(load_args.checkpoint_uid),
        )
        ts = load_args.timestep_to_restore
        restored = loader.restore_params(
            {"actor": learner.actor.state_dict(), "critic": learner.critic.state_dict()}, timestep=None if ts in (None, "null") else int(ts)
        )
        for _name, _mod in {"actor": learner.actor, "critic": learner.critic}.items():
            _mod.load_state_dict(restored[_name])
    lifetime = ThreadLifetime()
    pipeline = OnPolicyPipeline(n_actors)
    param_server = ParameterServer(n_actors)
    timers = TimingTracker()
    logger = StoixLogger(config) if ctx.is_main else None
    checkpointer = None
    if ctx.is_main and logger is not None and config.logger.checkpointing.save_model:
        from stoix_amd.utils.checkpointing import Checkpointer

        # reference Sebulba parity: the async evaluator saves the evaluated
        # snapshot per eval, best-by-return retained
        checkpointer = Checkpointer(
            model_name=config.system.system_name,
            metadata=config.to_plain() if hasattr(config, "to_plain") else dict(config),
            directory=f"{logger.directory}/checkpoints",
            max_to_keep=config.logger.checkpointing.save_args.max_to_keep,
            keep_period=config.logger.checkpointing.save_args.keep_period,
        )


    eval_env = env_factory(int(config.arch.num_eval_episodes))
    eval_actor = build_actor(config.network.actor_network, obs_space, act_space).to(eval_device)
    eval_gen = torch.Generator(device=eval_device)
    eval_gen.manual_seed(int(config.arch.seed) + 99_999)

    from stoix_amd.utils.sebulba import prewarm_convs

    prewarm_convs(config, learner.actor, learner.critic, obs_space.shape, learner_device)
    import os as _os

    torch.set_num_threads(min(8, _os.cpu_count() or 8))

    def evaluate_snapshot(params: Dict, t_env: int) -> Dict:
        from stoix_amd.evaluator import evaluate

        eval_actor.load_state_dict(params["actor"])

        def act(obs, greedy):
            dist = eval_actor(obs.to(eval_device))
            return (dist.mode() if greedy else dist.sample(eval_gen)).cpu()

        m = evaluate(act, eval_env, greedy=bool(config.arch.evaluation_greedy))
        if logger is not None:
            logger.log(m, t_env, 0, LogEvent.EVAL)
        if checkpointer is not None:
            checkpointer.save(
                t_env, params, metric_value=float(m["episode_return"].mean())
            )
        return m


    def absolute_snapshot_eval(params: Dict) -> Dict:
        from stoix_amd.evaluator import evaluate

        eval_actor.load_state_dict(params["actor"])

        def act(obs, greedy):
            dist = eval_actor(obs.to(eval_device))
            return (dist.mode() if greedy else dist.sample(eval_gen)).cpu()

        outs = [
            evaluate(act, eval_env, greedy=bool(config.arch.evaluation_greedy))
            for _ in range(10)
        ]
        return {
            k: torch.cat([o[k].reshape(-1) for o in outs]) for k in outs[0]
        }

    async_eval = AsyncEvaluator(evaluate_snapshot, lifetime)

    num_updates = int(config.arch.num_updates)
    # publish the INITIAL learner params before any actor starts: rollout 0
    # then acts with the learner's weights (matters after a load_model
    # restore; otherwise actors would spend rollout 0 on their own init)
    param_server.distribute_params(learner.cpu_params())
    threads = []
    for a_id in range(n_actors):
        dev = _device_of(actor_device_ids[a_id // per_dev], force_cpu)
        th = threading.Thread(
            target=actor_thread_fn,
            args=(a_id, config, env_factory, dev, pipeline, param_server, lifetime, num_updates, timers),
            daemon=True,
            name=f"actor-{a_id}",
        )
        th.start()
        threads.append(th)

    steps_per_update = int(config.system.rollout_length) * int(config.arch.total_num_envs)
    eval_every = max(1, num_updates // int(config.arch.num_evaluation))
    t_env = 0
    t0 = time.perf_counter()
    final_return = 0.0
    for update in range(num_updates):
        payloads = pipeline.collect_rollouts(lifetime)
        if payloads is None:
            break
        train_metrics = learner.learn(payloads)
        param_server.distribute_params(learner.cpu_params())
        t_env += steps_per_update
        if (update + 1) % eval_every == 0 and ctx.is_main:
            sps = t_env / (time.perf_counter() - t0)
            logger.log({"steps_per_second": sps, **train_metrics, **timers.summary()}, t_env, update, LogEvent.TRAIN)
            async_eval.submit_evaluation(learner.cpu_params(), t_env)

    # train-phase throughput (same contract as sebulba_ff_ppo: recorded
    # before the evaluator drain)
    run_experiment.last_sps = t_env / max(time.perf_counter() - t0, 1e-9)

    lifetime.stop()
    for th in threads:
        th.join(timeout=10)
    async_eval.join()
    if async_eval.last_metrics:
        final_return = float(async_eval.last_metrics["episode_return"].mean())
    # absolute metric: 10x episodes with the BEST evaluated params
    # (reference sebulba ff_ppo.py:994-1012)
    if (
        ctx.is_main
        and bool(getattr(config.arch, "absolute_metric", True))
        and async_eval.best_params is not None
    ):
        abs_m = absolute_snapshot_eval(async_eval.best_params)
        if logger is not None:
            logger.log(abs_m, t_env, int(config.arch.num_evaluation), LogEvent.ABSOLUTE)
        final_return = float(abs_m["episode_return"].mean())
    if logger is not None:
        logger.close()
    return final_return


def run(config) -> float:
    return run_experiment(config)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/sebulba/default_ff_impala.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
