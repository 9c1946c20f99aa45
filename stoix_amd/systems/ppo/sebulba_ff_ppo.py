"""Sebulba PPO — async actor threads with CPU envs feeding a GPU learner.

Parity with /root/reference/stoix/systems/ppo/sebulba/ff_ppo.py (call stack
SURVEY.md §3.2): N actor threads each run a fresh stateful vec env from the
env factory and a local policy copy on their actor device; rollouts (with a
T+1 value trace — GAE runs on the trace, no re-evaluation, ff_ppo.py:68-78,
396-411) flow through the bounded OnPolicyPipeline to the learner thread;
params return via the ParameterServer (actors fetch per rollout except the
first — one-step-off-policy pipelining, :205-213; ``arch.synchronous``
forces a blocking fetch); an AsyncEvaluator thread scores snapshots.

MI355X design: one process drives one learner GPU; the actor->learner
trajectory transfer goes through pinned buffers on a side HIP stream
(PinnedCopier). Multi-GPU learners = torchrun ranks with the same RCCL
flat-grad all-reduce as Anakin.
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
from stoix_amd.envs.env import get_final_step_metrics
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.ops.losses import clipped_value_loss, ppo_clip_loss
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module, get_dist_context
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


def _device_of(device_id, force_cpu: bool) -> torch.device:
    if force_cpu or not torch.cuda.is_available():
        return torch.device("cpu")
    return torch.device("cuda", int(device_id) % max(1, torch.cuda.device_count()))


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
    actor_net = build_actor(config.network.actor_network, obs_space, act_space).to(actor_device)
    critic_net = build_critic(config.network.critic_network, obs_space).to(actor_device)
    gen = torch.Generator(device=actor_device)
    gen.manual_seed(int(config.arch.seed) * 100 + actor_id)
    T = int(config.system.rollout_length)
    synchronous = bool(getattr(config.arch, "synchronous", False))

    ts = env.reset()
    for rollout_idx in range(num_rollouts):
        if lifetime.should_stop():
            return
        # param fetch: blocking on rollout>0 only when synchronous
        params = param_server.get_params(actor_id, block=(rollout_idx > 0 and synchronous), timeout=5.0)
        if params is None and rollout_idx > 0:
            params = param_server.get_params(actor_id, block=False)
        if params is not None:
            actor_net.load_state_dict(params["actor"])
            critic_net.load_state_dict(params["critic"])

        obs_l, act_l, logp_l, val_l, rew_l, disc_l, trunc_l = [], [], [], [], [], [], []
        # pinned fast path: native-pool envs expose their pinned CURRENT-obs
        # buffer; transfer THAT for inference (DMA) and keep the clone the
        # env returned for the payload. bf16 autocast for the actor nets
        # (inference only; the learner trains its own copy).
        pinned_src = getattr(env, "_obs", None)
        use_pin = (
            actor_device.type == "cuda"
            and pinned_src is not None
            and pinned_src.is_pinned()
        )
        amp = (
            torch.autocast("cuda", torch.bfloat16)
            if actor_device.type == "cuda"
            else None
        )
        with torch.no_grad():
            for _ in range(T):
                if use_pin:
                    obs_dev = pinned_src.to(actor_device, non_blocking=True)
                else:
                    obs_dev = ts.observation.to(actor_device)
                with timers.time("inference"):
                    if amp is not None:
                        with amp:
                            dist = actor_net(obs_dev)
                            value = critic_net(obs_dev)
                    else:
                        dist = actor_net(obs_dev)
                        value = critic_net(obs_dev)
                    action = dist.sample(gen)
                    logp = dist.log_prob(action)
                cpu_action = action.cpu()
                with timers.time("env_step"):
                    next_ts = env.step(cpu_action)
                obs_l.append(ts.observation)
                act_l.append(cpu_action)
                logp_l.append(logp.float().cpu())
                val_l.append(value.float().cpu())
                rew_l.append(next_ts.reward)
                disc_l.append(next_ts.discount)
                trunc_l.append(next_ts.truncated())
                ts = next_ts
            # bootstrap value for the trace
            if amp is not None:
                with amp:
                    last_val = critic_net(ts.observation.to(actor_device)).float().cpu()
            else:
                last_val = critic_net(ts.observation.to(actor_device)).cpu()

        payload = {
            "obs": torch.stack(obs_l),
            "action": torch.stack(act_l),
            "log_prob": torch.stack(logp_l),
            "value": torch.stack(val_l),
            "reward": torch.stack(rew_l),
            "discount": torch.stack(disc_l),
            "truncated": torch.stack(trunc_l),
            "last_value": last_val,
            "episode_metrics": {k: v.clone() for k, v in ts.extras["episode_metrics"].items()},
        }
        with timers.time("pipeline_put"):
            pipeline.send_rollout(actor_id, payload, lifetime)


class SebulbaPPOLearner:
    def __init__(self, config, learner_device: torch.device, obs_space, act_space):
        self.cfg = config
        self.sys = config.system
        self.device = learner_device
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(learner_device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(learner_device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr), eps=1e-5)
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=float(self.sys.critic_lr), eps=1e-5)
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()), learner_device
        )
        self.gen = torch.Generator(device=learner_device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 811)
        self.copier = PinnedCopier(learner_device)

    def cpu_params(self) -> Dict[str, Dict[str, Tensor]]:
        return {
            "actor": {k: v.detach().cpu() for k, v in self.actor.state_dict().items()},
            "critic": {k: v.detach().cpu() for k, v in self.critic.state_dict().items()},
        }

    def learn(self, payloads: List[Dict[str, Tensor]]) -> Dict[str, Tensor]:
        sysc = self.sys
        # concat actor payloads along the env axis, move via pinned staging
        merged: Dict[str, Tensor] = {}
        for k in ("obs", "action", "log_prob", "value", "reward", "discount", "truncated", "last_value"):
            merged[k] = torch.cat([p[k] for p in payloads], dim=-1 if payloads[0][k].dim() == 1 else 1)
        batch = self.copier.to_device(merged)

        T = batch["obs"].shape[0]
        B = batch["obs"].shape[1]
        gamma = float(sysc.gamma)
        # GAE over the value trace: v_next[t] = value[t+1], last from trace
        v = batch["value"]
        v_next = torch.cat([v[1:], batch["last_value"].unsqueeze(0)], dim=0)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            batch["reward"],
            gamma * batch["discount"],
            float(sysc.gae_lambda),
            v,
            v_next,
            truncation_t=batch["truncated"].bool(),
            standardize_advantages=bool(sysc.standardize_advantages),
        )
        TB = T * B
        flat_obs = batch["obs"].reshape(TB, *batch["obs"].shape[2:])
        flat_act = batch["action"].reshape(TB, *batch["action"].shape[2:])
        flat_logp = batch["log_prob"].reshape(TB)
        flat_val = v.reshape(TB)
        flat_adv = adv.reshape(TB)
        flat_tgt = targets.reshape(TB)
        n_mb = int(sysc.num_minibatches)
        mb = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(sysc.epochs)):
            perm = torch.randperm(TB, device=self.device, generator=self.gen)
            for i in range(n_mb):
                idx = perm[i * mb : (i + 1) * mb]
                dist = self.actor(flat_obs[idx])
                new_logp = dist.log_prob(flat_act[idx])
                entropy = dist.entropy().mean()
                a_loss = ppo_clip_loss(new_logp, flat_logp[idx], flat_adv[idx], float(sysc.clip_eps))
                v_pred = self.critic(flat_obs[idx])
                v_loss = clipped_value_loss(v_pred, flat_val[idx], flat_tgt[idx], float(sysc.clip_eps))
                loss = a_loss - float(sysc.ent_coef) * entropy + float(sysc.vf_coef) * v_loss
                self.actor_opt.zero_grad(set_to_none=True)
                self.critic_opt.zero_grad(set_to_none=True)
                loss.backward()
                self.reducer.reduce()
                self.reducer.wait()
                nn.utils.clip_grad_norm_(self.actor.parameters(), float(sysc.max_grad_norm))
                nn.utils.clip_grad_norm_(self.critic.parameters(), float(sysc.max_grad_norm))
                self.actor_opt.step()
                self.critic_opt.step()
                metrics = {
                    "actor_loss": a_loss.detach(),
                    "value_loss": v_loss.detach(),
                    "entropy": entropy.detach(),
                }
        return metrics


def run_experiment(config, force_cpu: bool = False) -> float:
    ctx = get_dist_context(force_cpu=force_cpu)
    config.arch.n_devices = ctx.world_size
    check_total_timesteps(config)
    torch.manual_seed(int(config.arch.seed) + ctx.rank)
    # cap the intra-op pool: on big hosts (256 cores) torch defaults to
    # half the cores and the per-task overhead of batched env stepping
    # collapses (measured 23.8K SPS at 128 threads vs 503K at 4)
    import os as _os

    torch.set_num_threads(min(8, _os.cpu_count() or 8))
    # NOTE: the FIRST run on a fresh box pays MIOpen conv auto-tuning for
    # the CNN shapes (~minutes, one-off, cached afterwards): measured 8.8K
    # SPS on the tuning run vs 68-73K steady-state. MIOPEN_FIND_MODE=FAST
    # was tried and measured WORSE (fallback-path pathology on gfx950);
    # leave MIOpen's default find in place.

    actor_device_ids = list(config.arch.actor.device_ids)
    per_dev = int(config.arch.actor.actor_per_device)
    n_actors = len(actor_device_ids) * per_dev
    learner_device = _device_of(config.arch.learner.device_ids[0], force_cpu)
    eval_device = _device_of(getattr(config.arch, "evaluator_device_id", 0), force_cpu)

    env_factory = environments.make_factory(config, device="cpu")
    probe_env = env_factory(1)
    obs_space, act_space = probe_env.observation_space, probe_env.action_space

    learner = SebulbaPPOLearner(config, learner_device, obs_space, act_space)
    lifetime = ThreadLifetime()

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


    # evaluation: fresh eval env on the evaluator device's host side
    eval_env = env_factory(int(config.arch.num_eval_episodes))
    eval_actor = build_actor(config.network.actor_network, obs_space, act_space).to(eval_device)
    eval_gen = torch.Generator(device=eval_device)
    eval_gen.manual_seed(int(config.arch.seed) + 99_999)

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

    from stoix_amd.utils.sebulba import prewarm_convs

    prewarm_convs(config, learner.actor, learner.critic, obs_space.shape, learner_device)

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
        with timers.time("pipeline_collect"):
            payloads = pipeline.collect_rollouts(lifetime)
        if payloads is None:
            break
        with timers.time("learn"):
            train_metrics = learner.learn(payloads)
        param_server.distribute_params(learner.cpu_params())
        t_env += steps_per_update
        if (update + 1) % eval_every == 0 and ctx.is_main:
            elapsed = time.perf_counter() - t0
            sps = t_env / elapsed
            logger.log({"steps_per_second": sps, **train_metrics, **timers.summary()}, t_env, update, LogEvent.TRAIN)
            em = payloads[0]["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                logger.log({k: v.mean() for k, v in final.items()}, t_env, update, LogEvent.ACT)
            async_eval.submit_evaluation(learner.cpu_params(), t_env)

    # train-phase throughput, recorded BEFORE the evaluator drain: the
    # async evaluator plays full episodes to completion at shutdown, and a
    # better policy means a LONGER drain — wall-clock around the whole
    # experiment anti-correlates with training quality (bench/probe read
    # this attribute instead)
    train_elapsed = time.perf_counter() - t0
    run_experiment.last_sps = t_env / max(train_elapsed, 1e-9)
    run_experiment.last_env_steps = t_env

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
    cfg = compose("default/sebulba/default_ff_ppo.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
