#!/usr/bin/env python3
"""Flagship benchmark: Anakin PPO on the Ant-class physics env.

Driver contract (BASELINE.json): measures env steps/sec for the whole job —
Anakin PPO, Ant (27-obs / 8-act, 4096 envs per GPU), bf16 network compute,
rollout_length 128, 4 epochs x 16 minibatches, data-parallel over RCCL/xGMI
for N > 1. One bench "step" = one full update step (rollout + GAE + PPO
update), i.e. rollout_length * num_envs env steps per GPU. The environment
is the simulation itself (synthetic physics, random-init weights — RL has no
dataset). `steps_per_second` definition matches the reference
(/root/reference/stoix/systems/ppo/anakin/ff_ppo.py:589-595).

Launch (multi-GPU, by the driver):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=8)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--num-envs", type=int, default=4096, help="envs per GPU")
    p.add_argument("--rollout-length", type=int, default=128)
    p.add_argument("--env", type=str, default="brax/ant")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--no-graph", action="store_true", help="disable hip-graph capture")
    args = p.parse_args()

    import torch.distributed as dist

    from stoix_amd.config import compose
    from stoix_amd.parallel.dist import get_dist_context
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps
    from stoix_amd import envs as environments

    ctx = get_dist_context()
    n_gpus = ctx.world_size
    device = ctx.device
    on_gpu = device.type == "cuda"
    dtype = args.dtype if on_gpu else "fp32"

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        [
            f"env={args.env}",
            f"arch.total_num_envs={args.num_envs * n_gpus}",
            "arch.total_timesteps=null",
            "arch.num_updates=1000000",
            "arch.num_evaluation=1",
            f"system.rollout_length={args.rollout_length}",
            "system.epochs=4",
            "system.num_minibatches=16",
            f"system.compute_dtype={dtype}",
            "logger.loggers=[]",
            "logger.checkpointing.save_model=false",
        ],
    )
    cfg.arch.n_devices = n_gpus
    check_total_timesteps(cfg)
    torch.manual_seed(42 + ctx.rank)

    num_envs = int(cfg.arch.num_envs)
    env = environments.make_single(cfg, num_envs, device, seed=42 + 31 * ctx.rank)
    learner = PPOLearner(cfg, env, device)
    if on_gpu and not args.no_graph:
        try:
            from stoix_amd.ops.graph import try_enable_graphs

            try_enable_graphs(learner)
        except Exception as e:
            print(f"[bench] graph capture unavailable: {e}", file=sys.stderr)

    def barrier_sync() -> None:
        if ctx.initialized:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        learner.update_step()
    barrier_sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        learner.update_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if ctx.initialized:
        e = torch.tensor([elapsed], device=device if on_gpu else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    env_steps = args.steps * int(cfg.system.rollout_length) * num_envs * n_gpus
    value = env_steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if ctx.is_main:
        print(
            json.dumps(
                {
                    "metric": "env steps/sec (whole node), Anakin PPO on Brax Ant at 1/2/4/8 MI355X",
                    "value": value,
                    "unit": "env_steps/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": dtype,
                    "data": "synthetic (vectorised Ant-class physics env, random-init weights)",
                    "config": {
                        "model": "ppo_mlp_256x2_tanh_normal",
                        "global_batch": num_envs * n_gpus,
                        "seq_len": int(cfg.system.rollout_length),
                        "parallelism": f"dp{n_gpus}",
                        "env": args.env,
                        "epochs": 4,
                        "num_minibatches": 16,
                    },
                }
            )
        )


if __name__ == "__main__":
    main()
