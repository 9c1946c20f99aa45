"""Time the PPO phases (rollout / perm / epoch) separately, eager vs graph.
Run on a GPU box: python tools/phase_timing.py [--num-envs N] [--rollout T]"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num-envs", type=int, default=4096)
    p.add_argument("--rollout", type=int, default=128)
    args = p.parse_args()

    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd import envs as environments
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        [
            "env=brax/ant",
            f"arch.total_num_envs={args.num_envs}",
            "arch.total_timesteps=null",
            "arch.num_updates=100",
            "arch.num_evaluation=1",
            f"system.rollout_length={args.rollout}",
            "system.epochs=4",
            "system.num_minibatches=16",
            "system.compute_dtype=bf16",
            "logger.loggers=[]",
        ],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    env = environments.make_single(cfg, args.num_envs, device, seed=0)
    learner = PPOLearner(cfg, env, device)

    t_roll = timeit(learner.rollout_phase)
    t_perm = timeit(learner._new_perm)
    t_epoch = timeit(learner.epoch_phase)
    print(f"EAGER  rollout={t_roll:8.2f} ms  perm={t_perm:6.2f} ms  epoch={t_epoch:8.2f} ms  "
          f"update={t_roll + 4 * (t_perm + t_epoch):8.2f} ms")

    # per-env-step cost of the rollout
    print(f"EAGER  per-env-step rollout cost: {t_roll / args.rollout * 1000:7.1f} us")

    from stoix_amd.ops.graph import try_enable_graphs

    ok = try_enable_graphs(learner)
    print(f"graph capture: {ok}")
    g_roll, g_epoch = learner._graphs
    t_roll_g = timeit(lambda: g_roll.replay())
    t_epoch_g = timeit(lambda: g_epoch.replay())
    t_update = timeit(learner.update_step)
    print(f"GRAPH  rollout={t_roll_g:8.2f} ms  epoch={t_epoch_g:8.2f} ms  update={t_update:8.2f} ms")
    sps = args.rollout * args.num_envs / (t_update / 1000.0)
    print(f"GRAPH  env steps/s: {sps:,.0f}")


if __name__ == "__main__":
    main()
