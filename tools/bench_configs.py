"""SUPERSEDED (round 2): every BASELINE config is now driver-runnable
through the repo-root bench.py contract —
    python bench.py --config {cartpole,ppo,sac,sebulba,rainbow}
(one JSON line each, warmup + barrier-bracketed timing).
This tool is kept for the round-1 measurement provenance.

Measure the non-flagship BASELINE.json configs (steps/sec) on one GPU.

The driver's bench.py covers config #2 (Anakin PPO / Ant). This tool times
configs #1, #3, #4, #5 with the same steps_per_second definition
(env steps per wall-clock second around the learn call) and prints one JSON
line per config. Run on a GPU box:
    python tools/bench_configs.py [--quick]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def time_anakin(learner, T, B, steps, warmup):
    dev = learner.device
    for _ in range(warmup):
        learner.update_step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        learner.update_step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return (T * B * steps) / dt, dt / steps * 1e3


def bench_ppo_cartpole_cpu(quick):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=4", "arch.total_timesteps=null",
         "arch.num_updates=100", "arch.num_evaluation=1",
         "system.rollout_length=128", "system.num_minibatches=2", "system.epochs=4",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cpu")
    env = environments.make_single(cfg, 4, device, seed=0)
    learner = PPOLearner(cfg, env, device)
    sps, ms = time_anakin(learner, 128, 4, 3 if quick else 10, 1)
    return {"config": "anakin_ppo_cartpole_cpu_4envs", "steps_per_second": sps,
            "ms_per_update": ms, "device": "cpu"}


def bench_sac_humanoid(quick):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.sac.ff_sac import SACLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    B = 2048
    cfg = compose(
        "default/anakin/default_ff_sac.yaml",
        [f"env=brax/humanoid", f"arch.total_num_envs={B}", "arch.total_timesteps=null",
         "arch.num_updates=100", "arch.num_evaluation=1",
         "system.rollout_length=8", "system.epochs=8", f"system.batch_size={B}",
         "system.buffer_size=4000000", "system.warmup_steps=32",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    env = environments.make_single(cfg, B, device, seed=0)
    learner = SACLearner(cfg, env, device)
    if device.type == "cuda":
        from stoix_amd.ops.graph import try_enable_update_graph

        try_enable_update_graph(learner)
    sps, ms = time_anakin(learner, 8, B, 5 if quick else 20, 2)
    # HBM residency of the replay buffer
    buf_bytes = sum(v.numel() * v.element_size() for v in learner.buffer.storage.values()) \
        if hasattr(learner.buffer, "storage") else 0
    return {"config": "anakin_sac_humanoid_hbm_replay", "steps_per_second": sps,
            "ms_per_update": ms, "envs": B, "replay_bytes": buf_bytes,
            "device": str(device)}


def bench_rainbow_snake(quick):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.q_learning.ff_rainbow import RainbowLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    B = 1024
    cfg = compose(
        "default/anakin/default_ff_rainbow.yaml",
        [f"env=jumanji/snake", f"arch.total_num_envs={B}", "arch.total_timesteps=null",
         "arch.num_updates=100", "arch.num_evaluation=1",
         "system.rollout_length=4", "system.epochs=2", f"system.batch_size={B}",
         "system.buffer_size=500000", "system.warmup_steps=32", "system.n_step=3",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    env = environments.make_single(cfg, B, device, seed=0)
    learner = RainbowLearner(cfg, env, device)
    sps, ms = time_anakin(learner, 4, B, 5 if quick else 20, 2)
    return {"config": "anakin_rainbow_snake_per", "steps_per_second": sps,
            "ms_per_update": ms, "envs": B, "device": str(device)}


def bench_sebulba_breakout(quick):
    """Sebulba PPO on the Breakout pixel env: CPU actor threads feeding the
    GPU learner. Reports the end-to-end SPS over the whole run."""
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.sebulba_ff_ppo import run_experiment

    # operating point swept on MI355X: 256 envs x 4 actor threads = 9.0K
    # SPS; 512 x 8 drops to 5.1K (python-thread thrash), 32 x 2 to 1.2K
    # (startup-dominated)
    n_updates = 4 if quick else 24
    n_envs = 32 if quick else 256
    rollout = 32 if quick else 64
    actors = 2 if quick else 4
    cfg = compose(
        "default/sebulba/default_ff_ppo.yaml",
        ["env=envpool/breakout", "network=cnn",
         f"arch.total_num_envs={n_envs}", "arch.total_timesteps=null",
         f"arch.num_updates={n_updates}", "arch.num_evaluation=1",
         "arch.num_eval_episodes=4", f"arch.actor.actor_per_device={actors}",
         f"system.rollout_length={rollout}", "system.num_minibatches=2", "system.epochs=1",
         "logger.loggers=[]", "logger.checkpointing.save_model=false"],
    )
    t0 = time.perf_counter()
    run_experiment(cfg)
    dt = time.perf_counter() - t0
    steps = n_envs * rollout * n_updates
    return {"config": "sebulba_ppo_breakout_pixels", "steps_per_second": steps / dt,
            "wall_s": dt, "envs": n_envs, "actors": actors,
            "device": "cpu-actors + learner device"}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--quick", action="store_true")
    p.add_argument("--only", type=str, default="")
    args = p.parse_args()
    benches = {
        "cartpole": bench_ppo_cartpole_cpu,
        "sac": bench_sac_humanoid,
        "rainbow": bench_rainbow_snake,
        "sebulba": bench_sebulba_breakout,
    }
    for name, fn in benches.items():
        if args.only and name not in args.only:
            continue
        try:
            r = fn(args.quick)
            print(json.dumps(r))
        except Exception as e:
            print(json.dumps({"config": name, "error": repr(e)}))


if __name__ == "__main__":
    main()
