"""Isolate the GPU CartPole learning collapse: same learner, HIP env kernel
vs torch _step_fn fallback (both on GPU)."""
import json, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from stoix_amd import envs as environments
from stoix_amd.config import compose
from stoix_amd.systems.ppo.ff_ppo import PPOLearner
from stoix_amd.utils.total_timestep_checker import check_total_timesteps

def run(hip: bool):
    cfg = compose("default/anakin/default_ff_ppo.yaml", [
        "env=classic/cartpole","arch.total_num_envs=256","arch.total_timesteps=null",
        "arch.num_updates=60","arch.num_evaluation=1",
        "system.rollout_length=128","system.num_minibatches=8","system.epochs=4",
        "logger.loggers=[]"])
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(3)
    env = environments.make_single(cfg, 256, "cuda:0", seed=3)
    if not hip:
        env._hip = None  # force the torch _step_fn path on GPU tensors
    L = PPOLearner(cfg, env, torch.device("cuda:0"))
    curve = []
    for u in range(60):
        L.update_step()
        if (u+1) % 6 == 0:
            curve.append(round(float(L.episode_metrics.get("episode_return", torch.tensor(float("nan")))),1))
    print(json.dumps({"hip_env": hip, "curve": curve}))

run(False)
run(True)
