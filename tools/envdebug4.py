import json, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from stoix_amd import envs as environments
from stoix_amd.config import compose
from stoix_amd.systems.ppo.ff_ppo import PPOLearner
from stoix_amd.utils.total_timestep_checker import check_total_timesteps

def run(hip, seed):
    cfg = compose("default/anakin/default_ff_ppo.yaml", [
        "env=classic/cartpole","arch.total_num_envs=256","arch.total_timesteps=null",
        "arch.num_updates=48","arch.num_evaluation=1",f"arch.seed={seed}",
        "system.rollout_length=128","system.num_minibatches=8","system.epochs=4",
        "logger.loggers=[]"])
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(seed)
    env = environments.make_single(cfg, 256, "cuda:0", seed=seed)
    if not hip:
        env._hip = None
    L = PPOLearner(cfg, env, torch.device("cuda:0"))
    curve = []
    for u in range(48):
        m = L.update_step()
        if (u+1) % 8 == 0:
            r = float(L.episode_metrics.get("episode_return", torch.tensor(float("nan"))))
            curve.append((round(r,1), round(float(m["entropy"]),3), round(float(m["value_loss"]),1)))
    print(json.dumps({"hip": hip, "seed": seed, "ret_ent_vloss": curve}))

for seed in (1, 2, 3):
    run(False, seed)
    run(True, seed)
