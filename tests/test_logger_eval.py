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


def test_logger_concurrent_threads_no_interleaving(tmp_path):
    """Race gate for the logger facade (SURVEY §5.2 posture): 8 threads log
    concurrently into the json + csv sinks; every emitted line must be a
    complete, parseable record and none may be lost."""
    import json as _json
    import threading

    from stoix_amd.config import DotDict
    from stoix_amd.utils.logger import LogEvent, StoixLogger

    cfg = DotDict.wrap(
        {
            "loggers": ["json", "csv"],
            "base_exp_path": str(tmp_path),
            "system_name": "stress",
            "run_name": "race",
        }
    )
    logger = StoixLogger(cfg)
    N_THREADS, N_LOGS = 8, 50
    errs = []

    def worker(k):
        try:
            for i in range(N_LOGS):
                logger.log({"episode_return": float(k * 1000 + i)}, t=i, t_eval=0,
                           event=LogEvent.ACT)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(k,)) for k in range(N_THREADS)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    logger.close()
    assert not errs
    jpath = tmp_path / "stress" / "race" / "json"
    files = list(jpath.glob("*.json*"))
    assert files, list(tmp_path.rglob("*"))
    lines = files[0].read_text().strip().splitlines()
    # ndjson-style sinks: every line parses, and all 400 records landed
    parsed = [_json.loads(l) for l in lines if l.strip().startswith("{")]
    assert len(parsed) == N_THREADS * N_LOGS
    seen = {int(p["episode_return"]) for p in parsed}
    assert len(seen) == N_THREADS * N_LOGS


def test_evaluate_masks_post_done_rewards_exactly():
    """evaluate() must accumulate each slot's reward ONLY until its first
    done (autoreset keeps the env running underneath): cross-checked
    against a hand-stepped env with identical seed and action stream."""
    import torch

    from stoix_amd.envs.classic import CartPole
    from stoix_amd.evaluator import evaluate

    B = 8

    def act_fn(obs, greedy):
        # deterministic position-dependent policy -> varied episode ends
        return (obs[:, 0] > 0).long()

    env = CartPole(B, seed=11)
    out = evaluate(act_fn, env, greedy=True)

    env2 = CartPole(B, seed=11)
    ts = env2.reset()
    ret = torch.zeros(B)
    length = torch.zeros(B)
    finished = torch.zeros(B, dtype=torch.bool)
    for _ in range(env2.max_episode_steps + 1):
        ts = env2.step(act_fn(ts.observation, True))
        active = ~finished
        ret += ts.reward * active
        length += active.float()
        finished |= ts.last()
        if bool(finished.all()):
            break
    torch.testing.assert_close(out["episode_return"], ret)
    torch.testing.assert_close(out["episode_length"], length)
    assert bool(finished.all())
