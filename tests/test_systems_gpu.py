"""GPU smoke tests for the non-flagship systems (the driver re-runs
`pytest -m gpu` on a real MI355X at round end; these pin the off-policy /
search tiers to the HIP env kernels + device-resident buffers)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")

TINY = [
    "arch.total_timesteps=null", "arch.num_updates=2", "arch.num_evaluation=1",
    "arch.num_eval_episodes=4", "logger.loggers=[]",
    "logger.checkpointing.save_model=false",
]


def _run(module, default, extra):
    import importlib

    from stoix_amd.config import compose

    mod = importlib.import_module(module)
    cfg = compose(default, TINY + extra)
    r = mod.run(cfg)
    assert r == r


@requires_gpu
def test_gpu_sac_humanoid():
    _run("stoix_amd.systems.sac.ff_sac", "default/anakin/default_ff_sac.yaml",
         ["env=brax/humanoid", "arch.total_num_envs=128",
          "system.rollout_length=4", "system.batch_size=128",
          "system.buffer_size=8192", "system.warmup_steps=16"])


@requires_gpu
def test_gpu_rainbow_snake():
    _run("stoix_amd.systems.q_learning.ff_rainbow",
         "default/anakin/default_ff_rainbow.yaml",
         ["env=jumanji/snake", "arch.total_num_envs=64",
          "system.rollout_length=4", "system.batch_size=64",
          "system.buffer_size=4096", "system.warmup_steps=16",
          "system.n_step=3", "system.epochs=2"])


@requires_gpu
def test_gpu_dqn_cartpole():
    _run("stoix_amd.systems.q_learning.ff_dqn",
         "default/anakin/default_ff_dqn.yaml",
         ["env=classic/cartpole", "arch.total_num_envs=128",
          "system.rollout_length=4", "system.batch_size=128",
          "system.buffer_size=8192", "system.warmup_steps=16"])


@requires_gpu
def test_gpu_td3_ant():
    _run("stoix_amd.systems.ddpg.ff_td3", "default/anakin/default_ff_td3.yaml",
         ["env=brax/ant", "arch.total_num_envs=128",
          "system.rollout_length=4", "system.batch_size=128",
          "system.buffer_size=8192", "system.warmup_steps=16"])


@requires_gpu
def test_gpu_az_cartpole():
    _run("stoix_amd.systems.search.ff_az", "default/anakin/default_ff_az.yaml",
         ["env=classic/cartpole", "arch.total_num_envs=32",
          "system.rollout_length=4", "system.num_simulations=6",
          "system.num_minibatches=2", "system.epochs=1"])


@requires_gpu
def test_gpu_sebulba_ppo_breakout():
    """Config #4 shape on one GPU: CPU Breakout-pixel actor threads feeding
    a CUDA learner (inference + V-trace-free PPO update on device)."""
    import importlib

    from stoix_amd.config import compose

    mod = importlib.import_module("stoix_amd.systems.ppo.sebulba_ff_ppo")
    cfg = compose(
        "default/sebulba/default_ff_ppo.yaml",
        [
            "env=envpool/breakout", "network=cnn",
            "arch.total_num_envs=8", "arch.total_timesteps=null",
            "arch.num_updates=2", "arch.num_evaluation=1",
            "arch.num_eval_episodes=2", "arch.absolute_metric=false",
            "arch.actor.actor_per_device=2",
            "system.rollout_length=8", "system.num_minibatches=2",
            "system.epochs=1", "logger.loggers=[]",
            "logger.checkpointing.save_model=false",
        ],
    )
    r = mod.run(cfg)
    assert r == r


@requires_gpu
def test_gpu_ppo_xland_goal_grid():
    """Round-2 env family on device: goal-conditioned gridworld steps as
    CUDA tensors under PPO."""
    _run("stoix_amd.systems.ppo.ff_ppo", "default/anakin/default_ff_ppo.yaml",
         ["env=xland_minigrid/goal_grid", "arch.total_num_envs=256",
          "system.rollout_length=8", "system.num_minibatches=2",
          "system.epochs=1"])


@requires_gpu
def test_gpu_ppo_crafting():
    _run("stoix_amd.systems.ppo.ff_ppo", "default/anakin/default_ff_ppo.yaml",
         ["env=craftax/crafting", "arch.total_num_envs=256",
          "system.rollout_length=8", "system.num_minibatches=2",
          "system.epochs=1"])


@requires_gpu
def test_gpu_rec_ppo_pomdp():
    _run("stoix_amd.systems.ppo.rec_ppo", "default/anakin/default_rec_ppo.yaml",
         ["env=popjym/stateless_cartpole", "arch.total_num_envs=128",
          "system.rollout_length=8", "system.num_minibatches=2",
          "system.epochs=1"])


@requires_gpu
def test_gpu_dqn_capture_on_snake_capture_safe_env():
    """The capture_safe torch-env path (no HIP kernel) must graph-capture a
    whole off-policy update end to end (widened gate, round 2)."""
    import torch as _t

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.ops.graph import try_enable_update_graph
    from stoix_amd.systems.q_learning.ff_dqn import DQNLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_dqn.yaml",
        ["env=jumanji/snake", "arch.total_num_envs=64",
         "arch.total_timesteps=null", "arch.num_updates=4",
         "arch.num_evaluation=1", "system.rollout_length=2",
         "system.batch_size=64", "system.buffer_size=4096",
         "system.warmup_steps=16", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    dev = _t.device("cuda:0")
    env = environments.make_single(cfg, 64, dev, seed=0)
    learner = DQNLearner(cfg, env, dev)
    assert learner.graph_capturable
    ok = try_enable_update_graph(learner)
    assert ok
    for _ in range(3):
        m = learner.update_step()
    _t.cuda.synchronize()
    assert _t.isfinite(m["q_loss"]).all()
