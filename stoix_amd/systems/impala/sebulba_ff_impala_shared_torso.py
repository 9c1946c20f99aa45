"""Sebulba IMPALA with a shared torso (parity:
/root/reference/stoix/systems/impala/sebulba/ff_impala_shared_torso.py).

Same V-trace learner as sebulba_ff_impala.py but with ONE network
(SharedPolicyValueNetwork: torso -> PolicyValueHead) and a single optimiser;
the actor threads run the same shared net for both the behaviour policy and
the value trace — one forward per step instead of two.
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
from stoix_amd.networks.factory import build_shared_policy_value
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module, get_dist_context
from stoix_amd.systems.ppo.sebulba_ff_ppo import _device_of
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


def actor_thread_fn(
    actor_id: int,
    config,
    env_factory,
    actor_device: torch.device,
    pipeline: OnPolicyPipeline,
    param_server: ParameterServer,
    lifetime: ThreadLifetime,
    num_rollouts: int,
    timers: TimingTracker,
) -> None:
    num_envs = int(config.arch.num_envs_per_actor)
    env = env_factory(num_envs)
    obs_space, act_space = env.observation_space, env.action_space
    net = build_shared_policy_value(config.network.actor_network, obs_space, act_space).to(
        actor_device
    )
    gen = torch.Generator(device=actor_device)
    gen.manual_seed(int(config.arch.seed) * 100 + actor_id)
    T = int(config.system.rollout_length)
    synchronous = bool(getattr(config.arch, "synchronous", False))

    ts = env.reset()
    for rollout_idx in range(num_rollouts):
        if lifetime.should_stop():
            return
        params = param_server.get_params(actor_id, block=(rollout_idx > 0 and synchronous), timeout=5.0)
        if params is None and rollout_idx > 0:
            params = param_server.get_params(actor_id, block=False)
        if params is not None:
            net.load_state_dict(params["net"])

        obs_l, act_l, logp_l, rew_l, disc_l = [], [], [], [], []
        with torch.no_grad():
            for _ in range(T):
                obs_dev = ts.observation.to(actor_device)
                with timers.time("inference"):
                    dist, _value = net(obs_dev)
                    action = dist.sample(gen)
                    logp = dist.log_prob(action)
                cpu_action = action.cpu()
                with timers.time("env_step"):
                    next_ts = env.step(cpu_action)
                obs_l.append(ts.observation)
                act_l.append(cpu_action)
                logp_l.append(logp.cpu())
                rew_l.append(next_ts.reward)
                disc_l.append(next_ts.discount)
                ts = next_ts
            _d, last_val = net(ts.observation.to(actor_device))

        payload = {
            "obs": torch.stack(obs_l),
            "action": torch.stack(act_l),
            "log_prob": torch.stack(logp_l),
            "reward": torch.stack(rew_l),
            "discount": torch.stack(disc_l),
            "last_value": last_val.cpu(),
            "episode_metrics": {k: v.clone() for k, v in ts.extras["episode_metrics"].items()},
        }
        with timers.time("pipeline_put"):
            pipeline.send_rollout(actor_id, payload, lifetime)


class SharedImpalaLearner:
    def __init__(self, config, device: torch.device, obs_space, act_space):
        self.cfg = config
        self.sys = config.system
        self.device = device
        self.net = build_shared_policy_value(config.network.actor_network, obs_space, act_space).to(
            device
        )
        broadcast_module(self.net)
        self.opt = torch.optim.Adam(self.net.parameters(), lr=float(self.sys.learner_lr), eps=1e-5)
        self.reducer = FlatGradReducer(list(self.net.parameters()), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 887)
        self.copier = PinnedCopier(device)

    def cpu_params(self) -> Dict[str, Dict[str, Tensor]]:
        return {"net": {k: v.detach().cpu() for k, v in self.net.state_dict().items()}}

    def learn(self, payloads: List[Dict[str, Tensor]]) -> Dict[str, Tensor]:
        sysc = self.sys
        merged: Dict[str, Tensor] = {}
        for k in ("obs", "action", "log_prob", "reward", "discount", "last_value"):
            merged[k] = torch.cat(
                [p[k] for p in payloads], dim=-1 if payloads[0][k].dim() == 1 else 1
            )
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
            dist, v_flat = self.net(flat_obs)
            logp = dist.log_prob(act.reshape(T * envs_per_mb, *act.shape[2:])).reshape(T, -1)
            entropy = dist.entropy().mean()
            v = v_flat.reshape(T, -1)
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
            loss = pg_loss + float(sysc.vf_coef) * critic_loss - float(sysc.ent_coef) * entropy
            self.opt.zero_grad(set_to_none=True)
            loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            nn.utils.clip_grad_norm_(self.net.parameters(), float(sysc.max_grad_norm))
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
    learner = SharedImpalaLearner(config, learner_device, obs_space, act_space)


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
            {"net": learner.net.state_dict()}, timestep=None if ts in (None, "null") else int(ts)
        )
        for _name, _mod in {"net": learner.net}.items():
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
    eval_net = build_shared_policy_value(config.network.actor_network, obs_space, act_space).to(
        eval_device
    )
    eval_gen = torch.Generator(device=eval_device)
    eval_gen.manual_seed(int(config.arch.seed) + 99_999)

    # MIOpen conv prewarm for the shared net's shapes (see
    # utils/sebulba.prewarm_convs; the shared-torso net returns
    # (dist, value) from one call so the helper's actor/critic split
    # doesn't apply)
    if learner_device.type == "cuda":
        with torch.random.fork_rng(devices=[learner_device]):
            mb_rows = max(
                1,
                int(config.system.rollout_length) * int(config.arch.total_num_envs)
                // int(config.system.num_minibatches),
            )
            with torch.no_grad(), torch.autocast("cuda", torch.bfloat16):
                for bs in {int(config.arch.num_envs_per_actor),
                           int(config.arch.num_eval_episodes)}:
                    learner.net(torch.zeros(bs, *obs_space.shape, device=learner_device))
            x = torch.zeros(mb_rows, *obs_space.shape, device=learner_device)
            d, v = learner.net(x)
            (d.entropy().sum() + v.sum()).backward()
            for p_ in learner.net.parameters():
                p_.grad = None
            del x, d, v
            torch.cuda.synchronize(learner_device)
    import os as _os

    torch.set_num_threads(min(8, _os.cpu_count() or 8))

    def evaluate_snapshot(params: Dict, t_env: int) -> Dict:
        from stoix_amd.evaluator import evaluate

        eval_net.load_state_dict(params["net"])

        def act(obs, greedy):
            dist, _v = eval_net(obs.to(eval_device))
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

        eval_net.load_state_dict(params["net"])

        def act(obs, greedy):
            dist, _v = eval_net(obs.to(eval_device))
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
    cfg = compose(
        "default/sebulba/default_ff_impala_shared_torso.yaml",
        argv if argv is not None else sys.argv[1:],
    )
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
