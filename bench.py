#!/usr/bin/env python3
"""Driver benchmark: the five BASELINE.json configs, one JSON line each.

Default (no flags): config #2 — the flagship Anakin PPO on the Ant-class
physics env (27-obs / 8-act, 4096 envs per GPU), bf16 network compute,
rollout_length 128, 4 epochs x 16 minibatches, data-parallel over RCCL/xGMI
for N > 1. One bench "step" = one full update step (rollout + GAE + PPO
update), i.e. rollout_length * num_envs env steps per GPU. The environment
is the simulation itself (synthetic physics, random-init weights — RL has
no dataset). `steps_per_second` matches the reference definition
(/root/reference/stoix/systems/ppo/anakin/ff_ppo.py:589-595).

--config selects the other BASELINE configs (driver-verifiable commands,
documented in BASELINE.md):
  ppo       Anakin PPO / Brax-Ant-class bf16, 4096 envs/GPU        (default)
  cartpole  Anakin PPO / CartPole, 4 envs (plumbing config)
  sac       Anakin SAC / Humanoid-class, HBM-resident replay
  rainbow   Anakin Rainbow-DQN / Snake, prioritised replay
  sebulba   Sebulba PPO / Breakout-class pixels, CPU envs -> GPU learner

Defaults (no flags): N=1, steps=60, warmup=5 — ~2 s of measured flagship
work on one MI355X, long enough for the driver's rocm-smi utilisation
sampler to catch >=2 samples (VERDICT r1 hygiene item).

Launch (multi-GPU, by the driver):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Multi-GPU first-try safety: the RCCL communicator is warmed with an eager
all-reduce before any graph capture; a failed graph-capture/replay falls
back to the eager path (STOIX_NO_GRAPH) instead of dying; per-rank device
pinning is asserted.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


# --------------------------------------------------------------- builders


def _anakin_learner(ctx, compose_args, learner_cls, num_envs, seed_mul=31):
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(*compose_args)
    cfg.arch.n_devices = ctx.world_size
    check_total_timesteps(cfg)
    torch.manual_seed(42 + ctx.rank)
    env = environments.make_single(cfg, num_envs, ctx.device, seed=42 + seed_mul * ctx.rank)
    learner = learner_cls(cfg, env, ctx.device)
    return cfg, learner


def build_ppo(args, ctx, dtype):
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner

    args.num_envs = args.num_envs or 4096
    n = ctx.world_size
    cfg, learner = _anakin_learner(
        ctx,
        (
            "default/anakin/default_ff_ppo_continuous.yaml",
            [
                f"env={args.env}",
                f"arch.total_num_envs={args.num_envs * n}",
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
        ),
        PPOLearner,
        args.num_envs,
    )
    meta = {
        "metric": "env steps/sec (whole node), Anakin PPO on Brax Ant at 1/2/4/8 MI355X",
        "steps_per_update": args.rollout_length * args.num_envs,
        "graph": "phases",
        "dtype": None,  # use CLI dtype (bf16 default)
        "config": {
            "model": "ppo_mlp_256x2_tanh_normal",
            "global_batch": args.num_envs * n,
            "seq_len": args.rollout_length,
            "parallelism": f"dp{n}",
            "env": args.env,
            "epochs": 4,
            "num_minibatches": 16,
        },
    }
    return learner, meta


def build_cartpole(args, ctx, dtype):
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner

    B = args.num_envs or 4
    n = ctx.world_size
    cfg, learner = _anakin_learner(
        ctx,
        (
            "default/anakin/default_ff_ppo.yaml",
            [
                "env=classic/cartpole",
                f"arch.total_num_envs={B * n}",
                "arch.total_timesteps=null",
                "arch.num_updates=1000000",
                "arch.num_evaluation=1",
                f"system.rollout_length={args.rollout_length}",
                "system.epochs=4",
                "system.num_minibatches=2",
                f"system.compute_dtype={dtype}",
                "logger.loggers=[]",
                "logger.checkpointing.save_model=false",
            ],
        ),
        PPOLearner,
        B,
    )
    meta = {
        "metric": "env steps/sec, Anakin PPO on CartPole (4 envs, plumbing config)",
        "steps_per_update": args.rollout_length * B,
        "graph": "phases",
        "dtype": None,
        "config": {
            "model": "ppo_mlp_256x2_categorical",
            "global_batch": B * n,
            "seq_len": args.rollout_length,
            "parallelism": f"dp{n}",
            "env": "classic/cartpole",
            "epochs": 4,
            "num_minibatches": 2,
        },
    }
    return learner, meta


def build_sac(args, ctx, dtype):
    from stoix_amd.systems.sac.ff_sac import SACLearner

    B = args.num_envs or 2048
    T = 8
    n = ctx.world_size
    cfg, learner = _anakin_learner(
        ctx,
        (
            "default/anakin/default_ff_sac.yaml",
            [
                "env=brax/humanoid",
                f"arch.total_num_envs={B * n}",
                "arch.total_timesteps=null",
                "arch.num_updates=1000000",
                "arch.num_evaluation=1",
                f"system.rollout_length={T}",
                "system.epochs=8",
                f"system.batch_size={B}",
                "system.buffer_size=4000000",
                "system.warmup_steps=32",
                "logger.loggers=[]",
                "logger.checkpointing.save_model=false",
            ],
        ),
        SACLearner,
        B,
    )
    meta = {
        "metric": "env steps/sec (whole node), Anakin SAC on Brax Humanoid, HBM-resident replay",
        "steps_per_update": T * B,
        "graph": "update",
        "dtype": "fp32",
        "config": {
            "model": "sac_twinq_mlp_256x2",
            "global_batch": B * n,
            "seq_len": T,
            "parallelism": f"dp{n}",
            "env": "brax/humanoid",
            "epochs": 8,
            "replay": "4M transitions HBM-resident",
        },
    }
    return learner, meta


def build_rainbow(args, ctx, dtype):
    from stoix_amd.systems.q_learning.ff_rainbow import RainbowLearner

    B = args.num_envs or 4096  # MI355X sizing; 1024 matches the round-1 table
    T = 4
    n = ctx.world_size
    cfg, learner = _anakin_learner(
        ctx,
        (
            "default/anakin/default_ff_rainbow.yaml",
            [
                "env=jumanji/snake",
                f"arch.total_num_envs={B * n}",
                "arch.total_timesteps=null",
                "arch.num_updates=1000000",
                "arch.num_evaluation=1",
                f"system.rollout_length={T}",
                "system.epochs=2",
                f"system.batch_size={B}",
                "system.buffer_size=500000",
                "system.warmup_steps=32",
                "system.n_step=3",
                f"system.compute_dtype={dtype}",
                "logger.loggers=[]",
                "logger.checkpointing.save_model=false",
            ],
        ),
        RainbowLearner,
        B,
    )
    meta = {
        "metric": "env steps/sec, Anakin Rainbow-DQN on Snake, prioritised replay",
        "steps_per_update": T * B,
        "graph": "update",
        "dtype": None,
        "config": {
            "model": "rainbow_noisy_dueling_c51",
            "global_batch": B * n,
            "seq_len": T,
            "parallelism": f"dp{n}",
            "env": "jumanji/snake",
            "epochs": 2,
            "replay": "prioritised sum-tree 500k",
        },
    }
    return learner, meta


BUILDERS = {
    "ppo": build_ppo,
    "cartpole": build_cartpole,
    "sac": build_sac,
    "rainbow": build_rainbow,
}


# ------------------------------------------------------------------ main


def bench_sebulba(args, ctx) -> None:
    """Config #4: Sebulba PPO, Breakout-class pixels, CPU env pool feeding
    the GPU learner through the pinned side-stream pipeline. Single
    process (actor threads + learner thread); a "step" here is one learner
    update (rollout_length * num_envs env steps)."""
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.sebulba_ff_ppo import run_experiment

    n_envs = args.num_envs or 2048
    rollout = min(args.rollout_length, 64)
    updates = args.steps + args.warmup
    actors = 8
    cfg = compose(
        "default/sebulba/default_ff_ppo.yaml",
        [
            "env=envpool/breakout",
            "network=cnn",
            f"arch.total_num_envs={n_envs}",
            "arch.total_timesteps=null",
            f"arch.num_updates={updates}",
            "arch.num_evaluation=1",
            "arch.num_eval_episodes=4",
            # no 10x absolute eval at shutdown: the bench measures the
            # train phase; last_sps is recorded before the drain either way
            "arch.absolute_metric=false",
            f"arch.actor.actor_per_device={actors}",
            f"system.rollout_length={rollout}",
            "system.num_minibatches=2",
            "system.epochs=1",
            "logger.loggers=[]",
            "logger.checkpointing.save_model=false",
        ],
    )
    t0 = time.perf_counter()
    run_experiment(cfg)
    dt = time.perf_counter() - t0
    env_steps = n_envs * rollout * updates
    # train-phase SPS (excludes the async-eval episode drain at shutdown,
    # whose length depends on policy quality, not engine speed)
    sps = getattr(run_experiment, "last_sps", env_steps / dt)
    print(
        json.dumps(
            {
                "metric": "env steps/sec, Sebulba PPO on Breakout-class pixels (CPU envs)",
                "value": sps,
                "unit": "env_steps/s",
                "n_gpus": 1,
                "steps": updates,
                "warmup": 0,
                "ms_per_step": (env_steps / sps) / updates * 1000.0,
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "fp32",
                "data": "synthetic (native C++ Breakout-class pixel pool, random-init weights)",
                "config": {
                    "model": "ppo_cnn",
                    "global_batch": n_envs,
                    "seq_len": rollout,
                    "parallelism": f"{actors}-actor-threads+1-learner",
                    "env": "envpool/breakout",
                },
                "note": "train-phase SPS (MIOpen prewarm + async-eval drain excluded)",
            }
        )
    )


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--config", type=str, default="ppo",
                   choices=["ppo", "cartpole", "sac", "rainbow", "sebulba"])
    p.add_argument("--num-envs", type=int, default=None, help="envs per GPU (default: per-config)")
    p.add_argument("--rollout-length", type=int, default=128)
    p.add_argument("--env", type=str, default="brax/ant")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--no-graph", action="store_true", help="disable hip-graph capture")
    args = p.parse_args()

    import torch.distributed as dist

    from stoix_amd.parallel.dist import get_dist_context

    ctx = get_dist_context()
    n_gpus = ctx.world_size
    device = ctx.device
    on_gpu = device.type == "cuda"
    dtype = args.dtype if on_gpu else "fp32"

    # per-rank device pinning: rank r must drive cuda:r (one process per GPU)
    if on_gpu and ctx.initialized:
        want = ctx.local_rank % torch.cuda.device_count()
        assert device.index == want, f"rank {ctx.rank}: device {device} != cuda:{want}"
        assert torch.cuda.current_device() == want

    # fail-fast RCCL/gloo connectivity check + communicator warmup BEFORE
    # any capture (the first collective builds the RCCL ring; building it
    # inside a hip-graph capture fails)
    if ctx.initialized:
        t = torch.ones(1, device=device if on_gpu else "cpu")
        dist.all_reduce(t)
        assert float(t.item()) == float(n_gpus), "all-reduce smoke check failed"
        if on_gpu:
            tb = torch.ones(8, dtype=torch.bfloat16, device=device)
            dist.all_reduce(tb)
            torch.cuda.synchronize(device)

    if args.config == "sebulba":
        if ctx.rank == 0:
            bench_sebulba(args, ctx)
        return

    def build(no_graph: bool):
        if no_graph:
            os.environ["STOIX_NO_GRAPH"] = "1"
        learner, meta = BUILDERS[args.config](args, ctx, dtype)
        if on_gpu and not (args.no_graph or no_graph):
            try:
                if meta["graph"] == "phases":
                    from stoix_amd.ops.graph import try_enable_graphs

                    try_enable_graphs(learner)
                else:
                    from stoix_amd.ops.graph import try_enable_update_graph

                    try_enable_update_graph(learner)
            except Exception as e:
                print(f"[bench] graph capture unavailable: {e!r}", file=sys.stderr)
        return learner, meta

    learner, meta = build(no_graph=False)
    # probe one update; if graph replay fails at world>1 (e.g. an RCCL-in-
    # graph issue on this stack), rebuild on the eager path rather than die
    try:
        learner.update_step()
        if on_gpu:
            torch.cuda.synchronize(device)
    except Exception as e:
        if getattr(learner, "_graphs", None) is None:
            raise
        print(f"[bench] graphed update failed ({e!r}); falling back to eager", file=sys.stderr)
        del learner
        if on_gpu:
            torch.cuda.empty_cache()
        learner, meta = build(no_graph=True)
        learner.update_step()
        if on_gpu:
            torch.cuda.synchronize(device)

    def barrier_sync() -> None:
        if ctx.initialized:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize(device)

    for _ in range(max(args.warmup - 1, 0)):  # probe counted as 1 warmup
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

    env_steps = args.steps * meta["steps_per_update"] * n_gpus
    value = env_steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if ctx.is_main:
        print(
            json.dumps(
                {
                    "metric": meta["metric"],
                    "value": value,
                    "unit": "env_steps/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": meta.get("dtype") or dtype,
                    "data": "synthetic (vectorised device-resident env, random-init weights)",
                    "config": meta["config"],
                }
            )
        )
    if ctx.initialized:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
