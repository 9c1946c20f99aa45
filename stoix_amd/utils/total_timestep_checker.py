"""Derived-quantity consistency checks for arch configs.

Parity with /root/reference/stoix/utils/total_timestep_checker.py
(anakin :9-131, sebulba :134-294): exactly one of
{total_timesteps, num_updates} may be None and is derived from the other;
division guards for envs/devices/batches.
"""
from __future__ import annotations

import torch


def check_total_timesteps(config) -> None:
    arch = config.arch
    system = config.system
    n_devices = int(getattr(arch, "n_devices", 0) or 0)
    if n_devices <= 0:
        n_devices = max(1, torch.cuda.device_count()) if torch.cuda.is_available() else 1
        arch.n_devices = n_devices

    if getattr(arch, "architecture_name", "anakin") == "anakin":
        update_batch_size = int(getattr(arch, "update_batch_size", 1))
        total_num_envs = int(arch.total_num_envs)
        assert total_num_envs % (n_devices * update_batch_size) == 0, (
            f"total_num_envs={total_num_envs} must divide by "
            f"n_devices*update_batch_size={n_devices * update_batch_size}"
        )
        arch.num_envs = total_num_envs // (n_devices * update_batch_size)
        steps_per_update = (
            int(system.rollout_length) * total_num_envs
        )
        if getattr(arch, "total_timesteps", None) is None:
            arch.total_timesteps = int(arch.num_updates) * steps_per_update
        else:
            arch.num_updates = max(1, int(arch.total_timesteps) // steps_per_update)
            derived = int(arch.num_updates) * steps_per_update
            if derived != int(arch.total_timesteps):
                print(
                    f"[timestep-checker] total_timesteps rounded "
                    f"{arch.total_timesteps} -> {derived} (steps/update={steps_per_update})"
                )
                arch.total_timesteps = derived
        num_eval = max(1, int(getattr(arch, "num_evaluation", 1)))
        assert int(arch.num_updates) >= num_eval, (
            f"num_updates={arch.num_updates} < num_evaluation={num_eval}"
        )
        arch.num_updates_per_eval = int(arch.num_updates) // num_eval
        if hasattr(system, "num_minibatches"):
            batch = int(arch.num_envs) * int(system.rollout_length)
            assert batch % int(system.num_minibatches) == 0, (
                f"rollout batch {batch} must divide num_minibatches={system.num_minibatches}"
            )
    else:  # sebulba
        total_num_envs = int(arch.total_num_envs)
        n_actors = len(arch.actor.device_ids) * int(arch.actor.actor_per_device)
        assert total_num_envs % n_actors == 0, (
            f"total_num_envs={total_num_envs} must divide by actor count {n_actors}"
        )
        arch.num_envs_per_actor = total_num_envs // n_actors
        n_learners = len(arch.learner.device_ids)
        assert arch.num_envs_per_actor % n_learners == 0 or n_learners == 1, (
            "envs per actor must shard evenly over learner devices"
        )
        steps_per_update = int(system.rollout_length) * total_num_envs
        if getattr(arch, "total_timesteps", None) is None:
            arch.total_timesteps = int(arch.num_updates) * steps_per_update
        else:
            arch.num_updates = max(1, int(arch.total_timesteps) // steps_per_update)
            arch.total_timesteps = int(arch.num_updates) * steps_per_update
        num_eval = max(1, int(getattr(arch, "num_evaluation", 1)))
        arch.num_updates_per_eval = int(arch.num_updates) // num_eval
