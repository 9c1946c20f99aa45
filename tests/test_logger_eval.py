"""Logger sinks + evaluator behaviors."""
import json
import os

import torch

from stoix_amd.config import compose
from stoix_amd.utils.logger import LogEvent, StoixLogger


def _cfg(tmp_path, loggers):
    return compose(
        "default/anakin/default_ff_ppo.yaml",
        [f"logger.loggers=[{','.join(loggers)}]",
         f"logger.base_exp_path={tmp_path}",
         "logger.checkpointing.save_model=false"],
    )


def test_logger_sinks_write(tmp_path):
    cfg = _cfg(tmp_path, ["json", "csv", "tensorboard"])
    lg = StoixLogger(cfg)
    lg.log({"episode_return": torch.tensor([1.0, 3.0])}, t=10, t_eval=0, event=LogEvent.EVAL)
    lg.log({"loss": 0.5}, t=10, t_eval=0, event=LogEvent.TRAIN)
    lg.close()
    files = [os.path.join(dp, f) for dp, _, fs in os.walk(tmp_path) for f in fs]
    exts = {os.path.splitext(f)[1] for f in files}
    assert ".jsonl" in exts and ".csv" in exts
    assert any("tfevents" in f for f in files)
    jl = [f for f in files if f.endswith(".jsonl")][0]
    rows = [json.loads(l) for l in open(jl)]
    ev = [r for r in rows if r["event"] == "evaluator"]
    assert ev and abs(ev[0]["episode_return"] - 2.0) < 1e-6  # described mean


def test_evaluator_solve_rate_and_absolute():
    from stoix_amd.envs.debug import IdentityGame
    from stoix_amd.evaluator import evaluate, evaluator_setup

    env = IdentityGame(num_envs=8, device="cpu", seed=0)
    env.solved_return_threshold = 0.0

    calls = {"n": 0}

    def act(obs, greedy):
        calls["n"] += 1
        return torch.randint(0, env.action_space.num_values, (obs.shape[0],))

    m = evaluate(act, env, greedy=True)
    assert "episode_return" in m and "episode_length" in m
    assert "solve_rate" in m

    class _Cfg:
        class arch:
            num_eval_episodes = 8
            evaluation_greedy = True

    eval_fn, abs_fn = evaluator_setup(env, _Cfg)
    r1 = eval_fn(act)
    r2 = abs_fn(act)
    assert r1["episode_return"].numel() <= r2["episode_return"].numel()
