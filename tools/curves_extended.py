"""Extended learning curves on GPU for the evidence pack
(profiles/r01_learning_curves.md): longer runs than tools/learncheck.py,
one JSON line per system. Run on a GPU box."""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def run_sac_humanoid(updates=3000):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.sac.ff_sac import SACLearner
    from stoix_amd.ops.graph import try_enable_update_graph
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    B = 1024
    cfg = compose(
        "default/anakin/default_ff_sac.yaml",
        ["env=brax/humanoid", f"arch.total_num_envs={B}", "arch.total_timesteps=null",
         f"arch.num_updates={updates}", "arch.num_evaluation=1",
         "system.rollout_length=8", "system.epochs=8", f"system.batch_size={B}",
         "system.buffer_size=1000000", "system.warmup_steps=32",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    torch.manual_seed(7)
    env = environments.make_single(cfg, B, device, seed=7)
    learner = SACLearner(cfg, env, device)
    try_enable_update_graph(learner)
    curve = []
    t0 = time.time()
    for u in range(updates):
        learner.update_step()
        if (u + 1) % max(1, updates // 12) == 0:
            learner.after_graph_replay()
            m = learner.episode_metrics
            curve.append(round(float(m.get("episode_return", torch.tensor(0.0)).float().mean()), 1))
    return {"system": "sac/humanoid", "curve": curve, "wall_s": round(time.time() - t0, 1)}


def run_rainbow_snake(updates=3000):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.q_learning.ff_rainbow import RainbowLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    B = 512
    cfg = compose(
        "default/anakin/default_ff_rainbow.yaml",
        ["env=jumanji/snake", f"arch.total_num_envs={B}", "arch.total_timesteps=null",
         f"arch.num_updates={updates}", "arch.num_evaluation=1",
         "system.rollout_length=4", "system.epochs=2", f"system.batch_size={B}",
         "system.buffer_size=200000", "system.warmup_steps=64", "system.n_step=3",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    torch.manual_seed(11)
    env = environments.make_single(cfg, B, device, seed=11)
    learner = RainbowLearner(cfg, env, device)
    curve = []
    t0 = time.time()
    for u in range(updates):
        learner.update_step()
        if (u + 1) % max(1, updates // 12) == 0:
            m = learner.episode_metrics
            curve.append(round(float(m.get("episode_return", torch.tensor(0.0)).float().mean()), 2))
    return {"system": "rainbow/snake", "curve": curve, "wall_s": round(time.time() - t0, 1)}


def run_ppo_ant_long(updates=1000):
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from learncheck import run_ppo

    return {"system": "ppo/ant(fused)", **run_ppo(
        "env=brax/ant", "cont", updates, 4096, ("system.compute_dtype=bf16",))}


def main():
    for fn in (run_ppo_ant_long, run_sac_humanoid, run_rainbow_snake):
        try:
            print(json.dumps(fn()), flush=True)
        except Exception as e:
            print(json.dumps({"system": fn.__name__, "error": repr(e)}), flush=True)


if __name__ == "__main__":
    main()
