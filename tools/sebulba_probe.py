"""Component-level Sebulba throughput probe.

Measures, separately: (a) raw native-pool stepping rate, (b) the actor
loop (inference + step, no pipeline), (c) the full experiment at several
operating points. Prints one JSON line per measurement.
"""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def probe_pool(B=256, steps=200):
    from stoix_amd.envs.envpool_cpu import BreakoutCpu

    env = BreakoutCpu(B, device="cpu", seed=0)
    env.reset()
    a = torch.randint(0, 4, (B,))
    for _ in range(10):
        env.step(a)
    t0 = time.perf_counter()
    for _ in range(steps):
        env.step(a)
    dt = time.perf_counter() - t0
    return {"probe": "pool_step", "B": B, "sps": B * steps / dt,
            "us_per_batch_step": dt / steps * 1e6}


def probe_actor_loop(B=256, T=64, rollouts=6, device="cuda:0"):
    from stoix_amd.config import compose
    from stoix_amd.envs.envpool_cpu import BreakoutCpu
    from stoix_amd.networks.factory import build_actor, build_critic

    cfg = compose("default/sebulba/default_ff_ppo.yaml",
                  ["env=envpool/breakout", "network=cnn"])
    env = BreakoutCpu(B, device="cpu", seed=0)
    dev = torch.device(device if torch.cuda.is_available() else "cpu")
    actor = build_actor(cfg.network.actor_network, env.observation_space, env.action_space).to(dev)
    critic = build_critic(cfg.network.critic_network, env.observation_space).to(dev)
    gen = torch.Generator(device=dev)
    ts = env.reset()
    t_inf = t_step = t_h2d = t_d2h = 0.0
    with torch.no_grad():
        for ro in range(rollouts):
            if ro == 1:
                t0 = time.perf_counter()
                t_inf = t_step = t_h2d = t_d2h = 0.0
            for _ in range(T):
                t = time.perf_counter()
                obs_dev = ts.observation.to(dev, non_blocking=False)
                if dev.type == "cuda":
                    torch.cuda.synchronize()
                t_h2d += time.perf_counter() - t
                t = time.perf_counter()
                dist = actor(obs_dev)
                value = critic(obs_dev)
                action = dist.sample(gen)
                if dev.type == "cuda":
                    torch.cuda.synchronize()
                t_inf += time.perf_counter() - t
                t = time.perf_counter()
                cpu_action = action.cpu()
                t_d2h += time.perf_counter() - t
                t = time.perf_counter()
                ts = env.step(cpu_action)
                t_step += time.perf_counter() - t
    dt = time.perf_counter() - t0
    n = B * T * (rollouts - 1)
    return {"probe": "actor_loop", "B": B, "T": T, "sps": n / dt,
            "split_ms_per_step": {"h2d": t_h2d / (T * (rollouts - 1)) * 1e3,
                                  "inference": t_inf / (T * (rollouts - 1)) * 1e3,
                                  "d2h": t_d2h / (T * (rollouts - 1)) * 1e3,
                                  "env": t_step / (T * (rollouts - 1)) * 1e3}}


def probe_e2e(n_envs, actors, rollout=64, updates=20):
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.sebulba_ff_ppo import run_experiment

    cfg = compose(
        "default/sebulba/default_ff_ppo.yaml",
        ["env=envpool/breakout", "network=cnn",
         f"arch.total_num_envs={n_envs}", "arch.total_timesteps=null",
         f"arch.num_updates={updates}", "arch.num_evaluation=1",
         "arch.num_eval_episodes=4", f"arch.actor.actor_per_device={actors}",
         f"system.rollout_length={rollout}", "system.num_minibatches=2",
         "system.epochs=1", "logger.loggers=[]",
         "logger.checkpointing.save_model=false"],
    )
    t0 = time.perf_counter()
    run_experiment(cfg)
    dt = time.perf_counter() - t0
    return {"probe": "e2e", "n_envs": n_envs, "actors": actors,
            "rollout": rollout,
            "sps": getattr(run_experiment, "last_sps", n_envs * rollout * updates / dt),
            "wall_s": dt}


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "pool"):
        for B in (256, 1024, 4096):
            print(json.dumps(probe_pool(B)))
    if which in ("all", "actor"):
        for B in (256, 1024):
            print(json.dumps(probe_actor_loop(B)))
    if which in ("all", "e2e"):
        for n_envs, actors in ((256, 4), (1024, 4), (2048, 8)):
            print(json.dumps(probe_e2e(n_envs, actors)))


def probe_impala(n_envs=2048, actors=8, rollout=64, updates=16):
    from stoix_amd.config import compose
    from stoix_amd.systems.impala.sebulba_ff_impala import run_experiment

    cfg = compose(
        "default/sebulba/default_ff_impala.yaml",
        ["env=envpool/breakout", "network=cnn",
         f"arch.total_num_envs={n_envs}", "arch.total_timesteps=null",
         f"arch.num_updates={updates}", "arch.num_evaluation=1",
         "arch.num_eval_episodes=4", f"arch.actor.actor_per_device={actors}",
         f"system.rollout_length={rollout}", "system.num_minibatches=2",
         "logger.loggers=[]", "logger.checkpointing.save_model=false"],
    )
    import time as _t

    t0 = _t.perf_counter()
    run_experiment(cfg)
    dt = _t.perf_counter() - t0
    return {"probe": "impala_e2e", "n_envs": n_envs, "actors": actors,
            "sps": getattr(run_experiment, "last_sps", n_envs * rollout * updates / dt),
            "wall_s": dt}
