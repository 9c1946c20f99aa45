"""Diagnose hip-graph capture failures for the Rainbow update path: run
the capture with a full traceback instead of the swallowed warning."""
import sys
import traceback

import torch

sys.path.insert(0, ".")

from stoix_amd import envs as environments
from stoix_amd.config import compose
from stoix_amd.ops.graph import try_enable_update_graph
from stoix_amd.utils.total_timestep_checker import check_total_timesteps
from stoix_amd.systems.q_learning.ff_rainbow import RainbowLearner

cfg = compose(
    "default/anakin/default_ff_rainbow.yaml",
    ["env=jumanji/snake", "arch.total_num_envs=64", "arch.total_timesteps=null",
     "arch.num_updates=8", "arch.num_evaluation=1", "system.rollout_length=4",
     "system.batch_size=64", "system.buffer_size=4096", "system.warmup_steps=16",
     "system.n_step=3", "system.epochs=2", "logger.loggers=[]"],
)
cfg.arch.n_devices = 1
check_total_timesteps(cfg)
dev = torch.device("cuda:0")
env = environments.make_single(cfg, 64, dev, seed=0)
learner = RainbowLearner(cfg, env, dev)
print("graph_capturable:", learner.graph_capturable)
try:
    ok = try_enable_update_graph(learner)
    print("capture ok:", ok)
    for _ in range(3):
        m = learner.update_step()
    torch.cuda.synchronize()
    print("replay ok, q_loss:", float(m["q_loss"]))
except Exception:
    traceback.print_exc()
