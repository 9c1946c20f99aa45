"""Anakin experiment runner — the shared host loop.

Structure parity with the reference's per-system ``run_experiment``
(/root/reference/stoix/systems/ppo/anakin/ff_ppo.py:554-706): the only
Python loop in training is over evaluation intervals; each interval runs
``num_updates_per_eval`` fused update steps on-device, then evaluates, logs
throughput (identical ``steps_per_second`` definition, ff_ppo.py:589-595),
checkpoints, and tracks best params for the absolute metric
(ff_ppo.py:673-699).

Learner objects implement:
  * ``update_step() -> Dict[str, Tensor]``  (one rollout + update, on-device)
  * ``act_fn`` / optional ``rnn_act`` for evaluation
  * ``state_for_checkpoint()``
Multi-GPU: one process per GPU (torchrun), RCCL all-reduce inside the
learner's optimiser step; rank 0 logs/evaluates.
"""
from __future__ import annotations

import time
from typing import Callable, Dict

import torch

from stoix_amd import envs as environments
from stoix_amd.evaluator import evaluator_setup
from stoix_amd.parallel.dist import get_dist_context
from stoix_amd.utils.checkpointing import Checkpointer
from stoix_amd.utils.logger import LogEvent, StoixLogger
from stoix_amd.utils.total_timestep_checker import check_total_timesteps


def default_aux_state(learner) -> dict:
    """Optimizer state_dicts for every torch Optimizer the learner holds
    (true-resume payload; learners can override with aux_checkpoint_state).
    The reference checkpoints the full LearnerState incl. opt_states."""
    return {
        name: opt.state_dict()
        for name, opt in vars(learner).items()
        if isinstance(opt, torch.optim.Optimizer)
    }


def load_default_aux_state(learner, aux: dict) -> None:
    """Counterpart of default_aux_state: restore optimizer moments by
    attribute name, ignoring entries the learner does not have."""
    for name, sd in (aux or {}).items():
        opt = getattr(learner, name, None)
        if isinstance(opt, torch.optim.Optimizer):
            opt.load_state_dict(sd)


def run_anakin_experiment(config, learner_factory: Callable, force_cpu: bool = False) -> float:
    ctx = get_dist_context(force_cpu=force_cpu)
    device = ctx.device
    config.arch.n_devices = ctx.world_size
    check_total_timesteps(config)
    torch.manual_seed(int(config.arch.seed) + ctx.rank)

    # update_batch_size folds exactly into the env batch: the reference's
    # vmapped independent learners stay bit-identical under per-minibatch
    # pmean, so N replicas == one learner with N x batch.
    num_envs = int(config.arch.num_envs) * int(getattr(config.arch, "update_batch_size", 1))
    train_env = environments.make_single(config, num_envs, device, int(config.arch.seed) + 31 * ctx.rank)
    eval_env = environments.make_single(
        config, int(config.arch.num_eval_episodes), device, int(config.arch.seed) + 10_000
    )

    learner = learner_factory(config, train_env, device)
    # search-based systems (AZ / sampled-AZ / SPO) evaluate with search
    # acting when an env handle is available (reference
    # stoix/systems/search/evaluator.py); plain systems ignore this.
    learner._eval_env_ref = eval_env

    # hip-graph capture of the update loop (rollout graph + epoch graph)
    # for capture-shaped learners on GPU (ops/graph.py); eager otherwise
    if device.type == "cuda" and hasattr(learner, "rollout_phase") and hasattr(learner, "epoch_phase"):
        try:
            from stoix_amd.ops.graph import try_enable_graphs

            try_enable_graphs(learner)
        except Exception as e:
            print(f"[anakin] hip-graph capture unavailable ({e!r}); running eager")
    elif device.type == "cuda" and getattr(learner, "graph_capturable", False):
        try:
            from stoix_amd.ops.graph import try_enable_update_graph

            try_enable_update_graph(learner)
        except Exception as e:
            print(f"[anakin] update-graph capture unavailable ({e!r}); running eager")

    eval_fn, absolute_eval_fn = evaluator_setup(eval_env, config)
    logger = StoixLogger(config) if ctx.is_main else None
    checkpointer = None
    if ctx.is_main and config.logger.checkpointing.save_model:
        checkpointer = Checkpointer(
            model_name=config.system.system_name,
            metadata=config.to_plain() if hasattr(config, "to_plain") else dict(config),
            directory=f"{logger.directory}/checkpoints",
            max_to_keep=config.logger.checkpointing.save_args.max_to_keep,
            keep_period=config.logger.checkpointing.save_args.keep_period,
        )

    # Restore-at-startup (reference ff_ppo.py:504-512: load_model rebuilds
    # params from a saved checkpoint before training). checkpoint_uid is
    # the DIRECTORY holding <system_name>/step_*/; optimizer moments are
    # restored too when the checkpoint carries an aux payload.
    load_cfg = getattr(config.logger.checkpointing, "load_model", False)
    if load_cfg:
        load_args = config.logger.checkpointing.load_args
        load_dir = load_args.checkpoint_uid
        if not load_dir:
            raise ValueError(
                "logger.checkpointing.load_model=true needs "
                "logger.checkpointing.load_args.checkpoint_uid "
                "(the directory that contains <system_name>/step_*/)"
            )
        loader = Checkpointer(
            model_name=config.system.system_name, directory=str(load_dir)
        )
        ts_restore = load_args.timestep_to_restore
        restored = loader.restore_params(
            learner.state_for_checkpoint(),
            timestep=None if ts_restore in (None, "null") else int(ts_restore),
        )
        learner.load_params(restored)
        aux = loader.restore_aux(
            timestep=None if ts_restore in (None, "null") else int(ts_restore)
        )
        if aux is not None:
            aux_load = getattr(learner, "load_aux_checkpoint_state", None)
            if aux_load is not None:
                aux_load(aux)
            else:
                load_default_aux_state(learner, aux)

    num_updates_per_eval = int(config.arch.num_updates_per_eval)
    steps_per_update = (
        int(config.system.rollout_length) * num_envs * ctx.world_size
    )
    steps_per_eval_interval = steps_per_update * num_updates_per_eval

    best_return = float("-inf")
    best_state = None
    final_return = 0.0
    t_env = 0
    rnn = bool(getattr(learner, "is_recurrent", False))

    for eval_idx in range(int(config.arch.num_evaluation)):
        t0 = time.perf_counter()
        train_metrics: Dict[str, torch.Tensor] = {}
        for _ in range(num_updates_per_eval):
            train_metrics = learner.update_step()
        if device.type == "cuda":
            torch.cuda.synchronize(device)
        elapsed = time.perf_counter() - t0
        t_env += steps_per_eval_interval
        sps = steps_per_eval_interval / elapsed

        if ctx.is_main:
            logger.log({"steps_per_second": sps, **train_metrics}, t_env, eval_idx, LogEvent.TRAIN)
            ep_metrics = getattr(learner, "episode_metrics", None)
            if ep_metrics:
                ep_metrics = dict(ep_metrics)
                has_final = ep_metrics.pop("has_final", None)
                if has_final is None or bool(has_final):
                    logger.log(ep_metrics, t_env, eval_idx, LogEvent.ACT)

            eval_metrics = eval_fn(learner.act_fn, rnn=rnn)
            mean_return = float(eval_metrics["episode_return"].mean())
            final_return = mean_return
            logger.log(eval_metrics, t_env, eval_idx, LogEvent.EVAL)

            save_every = int(
                getattr(config.logger.checkpointing.save_args, "save_interval_steps", 1)
                or 1
            )
            if checkpointer is not None and eval_idx % save_every == 0:
                aux_fn = getattr(learner, "aux_checkpoint_state", None)
                aux = aux_fn() if aux_fn is not None else default_aux_state(learner)
                checkpointer.stage_aux(aux or None)
                checkpointer.save(t_env, learner.state_for_checkpoint(), mean_return)
            if mean_return >= best_return:
                best_return = mean_return
                best_state = learner.snapshot_params()

    if ctx.is_main and bool(getattr(config.arch, "absolute_metric", True)) and best_state is not None:
        learner.load_params(best_state)
        abs_metrics = absolute_eval_fn(learner.act_fn, rnn=rnn)
        logger.log(abs_metrics, t_env, int(config.arch.num_evaluation), LogEvent.ABSOLUTE)
        final_return = float(abs_metrics["episode_return"].mean())

    if ctx.is_main and logger is not None:
        logger.close()
    return final_return
