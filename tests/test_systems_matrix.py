"""Smoke matrix: every system end-to-end at tiny scale on CPU (parity with
the reference's bash_scripts/run-algorithms.sh)."""
import importlib

import pytest

from stoix_amd.config import compose
from stoix_amd.parallel.dist import reset_dist_context

TINY = [
    "arch.total_num_envs=8",
    "arch.total_timesteps=null",
    "arch.num_updates=2",
    "arch.num_evaluation=1",
    "arch.num_eval_episodes=4",
    "arch.absolute_metric=false",
    "system.rollout_length=4",
    "logger.loggers=[]",
    "logger.checkpointing.save_model=false",
]

SYSTEMS = [
    ("stoix_amd.systems.ppo.ff_ppo", "default/anakin/default_ff_ppo.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.ppo.ff_ppo", "default/anakin/default_ff_ppo_continuous.yaml",
     ["system.num_minibatches=2", "system.epochs=1", "env=classic/pendulum"]),
    ("stoix_amd.systems.ppo.ff_ppo_penalty", "default/anakin/default_ff_ppo_penalty.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.ppo.ff_ppo_penalty", "default/anakin/default_ff_ppo_penalty_continuous.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.ppo.ff_dpo", "default/anakin/default_ff_dpo_continuous.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.q_learning.ff_dqn", "default/anakin/default_ff_dqn.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_ddqn", "default/anakin/default_ff_ddqn.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_mdqn", "default/anakin/default_ff_mdqn.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_dqn_reg", "default/anakin/default_ff_dqn_reg.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_c51", "default/anakin/default_ff_c51.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_qr_dqn", "default/anakin/default_ff_qr_dqn.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.q_learning.ff_pqn", "default/anakin/default_ff_pqn.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.sac.ff_sac", "default/anakin/default_ff_sac.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.ddpg.ff_ddpg", "default/anakin/default_ff_ddpg.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.ddpg.ff_td3", "default/anakin/default_ff_td3.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.ddpg.ff_d4pg", "default/anakin/default_ff_d4pg.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16"]),
    ("stoix_amd.systems.vpg.ff_reinforce", "default/anakin/default_ff_reinforce.yaml", []),
    ("stoix_amd.systems.awr.ff_awr", "default/anakin/default_ff_awr.yaml",
     ["system.num_critic_steps=2", "system.num_actor_steps=2", "system.batch_size=8",
      "system.buffer_size=256", "system.sample_sequence_length=4"]),
    ("stoix_amd.systems.awr.ff_awr", "default/anakin/default_ff_awr_continuous.yaml",
     ["system.num_critic_steps=2", "system.num_actor_steps=2", "system.batch_size=8",
      "system.buffer_size=256", "system.sample_sequence_length=4"]),
    ("stoix_amd.systems.mpo.ff_mpo", "default/anakin/default_ff_mpo.yaml",
     ["system.epochs=2", "system.batch_size=8", "system.buffer_size=256",
      "system.sample_sequence_length=4", "system.num_samples=4"]),
    ("stoix_amd.systems.mpo.ff_mpo", "default/anakin/default_ff_mpo_continuous.yaml",
     ["system.epochs=2", "system.batch_size=8", "system.buffer_size=256",
      "system.sample_sequence_length=4", "system.num_samples=4"]),
    ("stoix_amd.systems.mpo.ff_vmpo", "default/anakin/default_ff_vmpo.yaml", []),
    ("stoix_amd.systems.mpo.ff_vmpo", "default/anakin/default_ff_vmpo_continuous.yaml", []),
    ("stoix_amd.systems.q_learning.ff_rainbow", "default/anakin/default_ff_rainbow.yaml",
     ["system.epochs=2", "system.batch_size=16", "system.buffer_size=512",
      "system.warmup_steps=16", "system.n_step=3"]),
    ("stoix_amd.systems.ppo.rec_ppo", "default/anakin/default_rec_ppo.yaml",
     ["system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.q_learning.rec_r2d2", "default/anakin/default_rec_r2d2.yaml",
     ["system.epochs=1", "system.batch_size=8", "system.buffer_size=512",
      "system.sample_sequence_length=8", "system.burn_in_length=2", "system.n_step=2",
      "system.rollout_length=10"]),
    ("stoix_amd.systems.search.ff_az", "default/anakin/default_ff_az.yaml",
     ["system.num_simulations=6", "system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.search.ff_mz", "default/anakin/default_ff_mz.yaml",
     ["system.num_simulations=4", "system.epochs=1", "system.unroll_steps=2",
      "system.n_step=2", "system.batch_size=8", "system.buffer_size=256",
      "system.rollout_length=8"]),
    ("stoix_amd.systems.search.ff_sampled_az", "default/anakin/default_ff_sampled_az.yaml",
     ["system.num_simulations=4", "system.num_sampled_actions=4",
      "system.num_minibatches=2", "system.epochs=1"]),
    ("stoix_amd.systems.spo.ff_spo", "default/anakin/default_ff_spo.yaml",
     ["system.num_particles=4", "system.search_depth=2", "system.num_minibatches=2",
      "system.epochs=1"]),
    ("stoix_amd.systems.spo.ff_spo", "default/anakin/default_ff_spo_continuous.yaml",
     ["system.num_particles=4", "system.search_depth=2", "system.num_minibatches=2",
      "system.epochs=1"]),
    ("stoix_amd.systems.search.ff_sampled_mz", "default/anakin/default_ff_sampled_mz.yaml",
     ["system.num_simulations=4", "system.num_sampled_actions=4", "system.epochs=1",
      "system.unroll_steps=2", "system.n_step=2", "system.batch_size=8",
      "system.buffer_size=256", "system.rollout_length=8"]),
]


@pytest.fixture(autouse=True)
def _fresh_dist():
    reset_dist_context()
    yield
    reset_dist_context()


@pytest.mark.parametrize("module,default,extra", SYSTEMS, ids=[f"{m.split('.')[-1]}-{d.split('_', 2)[-1][:-5]}" for m, d, e in SYSTEMS])
def test_system_smoke(module, default, extra):
    mod = importlib.import_module(module)
    cfg = compose(default, TINY + extra)
    r = mod.run(cfg)
    assert r == r, f"{module} returned NaN"


def test_disco103_smoke():
    """DisCo-103 runs end-to-end with the random-init meta-network
    fallback (reference ff_disco103.py; meta-params download is offline-
    gated — docstring of systems/disco_rl/ff_disco103.py)."""
    from stoix_amd.config import compose
    from stoix_amd.systems.disco_rl import ff_disco103

    cfg = compose(
        "default/anakin/default_ff_disco103.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=8",
         "arch.total_timesteps=null", "arch.num_updates=2",
         "arch.num_evaluation=1", "arch.num_eval_episodes=4",
         "system.rollout_length=8", "system.num_minibatches=2",
         "system.epochs=1", "system.disco_rule.num_bins=21",
         "system.disco_rule.max_abs_value=50.0",
         "system.disco_rule.net.prediction_size=16",
         "logger.loggers=[]", "logger.checkpointing.save_model=false"],
    )
    r = ff_disco103.run(cfg)
    assert r == r  # finite float out of the experiment
