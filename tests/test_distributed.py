"""Multi-process data-parallel tests (gloo backend, world_size=2, CPU).

The driver runs the 8-GPU scaling bench at round end; these tests pin the
same code paths on CPU: FlatGradReducer's fused mean all-reduce, init-time
parameter broadcast, and a full 2-rank eager PPO update keeping ranks
bit-identical after every step.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


def _init(rank: int, port: int):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["LOCAL_RANK"] = str(rank)
    from stoix_amd.parallel.dist import get_dist_context, reset_dist_context

    reset_dist_context()
    return get_dist_context(force_cpu=True)


def _reducer_worker(rank: int, port: int, outdir: str):
    try:
        ctx = _init(rank, port)
        from stoix_amd.parallel.dist import FlatGradReducer

        torch.manual_seed(100 + rank)  # DIFFERENT grads per rank
        lin = torch.nn.Linear(8, 4)
        loss = lin(torch.randn(16, 8)).pow(2).mean()
        loss.backward()
        grads_before = [p.grad.clone() for p in lin.parameters()]
        red = FlatGradReducer(list(lin.parameters()), ctx.device)
        red.reduce()
        red.wait()
        torch.save(([p.grad.clone() for p in lin.parameters()], grads_before),
                   os.path.join(outdir, f"r{rank}.pt"))
    except Exception as e:  # pragma: no cover
        torch.save(("ERROR", repr(e)), os.path.join(outdir, f"r{rank}.pt"))


def _ppo_worker(rank: int, port: int, outdir: str):
    try:
        _init(rank, port)
        from stoix_amd import envs as environments
        from stoix_amd.config import compose
        from stoix_amd.systems.ppo.ff_ppo import PPOLearner
        from stoix_amd.utils.total_timestep_checker import check_total_timesteps

        torch.manual_seed(7 + rank)  # different init per rank pre-broadcast
        cfg = compose(
            "default/anakin/default_ff_ppo.yaml",
            [
                "env=classic/cartpole", "arch.total_num_envs=8",
                "arch.total_timesteps=null", "arch.num_updates=2",
                "arch.num_evaluation=1", "arch.seed=5",
                "system.rollout_length=8", "system.num_minibatches=2",
                "system.epochs=1", "logger.loggers=[]",
            ],
        )
        cfg.arch.n_devices = WORLD
        check_total_timesteps(cfg)
        device = torch.device("cpu")
        env = environments.make_single(cfg, 4, device, seed=5 + 31 * rank)
        learner = PPOLearner(cfg, env, device)
        for _ in range(2):
            learner.update_step()
        snap = {k: v.clone() for k, v in learner.actor.state_dict().items()}
        torch.save((snap, None), os.path.join(outdir, f"r{rank}.pt"))
    except Exception as e:  # pragma: no cover
        torch.save(("ERROR", repr(e)), os.path.join(outdir, f"r{rank}.pt"))


def _run_workers(fn, port):
    import tempfile

    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as outdir:
        procs = [ctx.Process(target=fn, args=(r, port, outdir)) for r in range(WORLD)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
        results = {}
        for r in range(WORLD):
            path = os.path.join(outdir, f"r{r}.pt")
            assert os.path.exists(path), f"rank {r} produced no result"
            a, b = torch.load(path, weights_only=False)
            assert not (isinstance(a, str) and a == "ERROR"), f"rank {r}: {b}"
            results[r] = (a, b)
    return results


def test_flat_grad_reducer_means_across_ranks():
    res = _run_workers(_reducer_worker, 29611)
    g0_after, g0_before = res[0]
    g1_after, g1_before = res[1]
    for a0, a1, b0, b1 in zip(g0_after, g1_after, g0_before, g1_before):
        torch.testing.assert_close(a0, a1)  # ranks agree after reduce
        torch.testing.assert_close(a0, (b0 + b1) / 2, rtol=1e-6, atol=1e-7)


def test_two_rank_ppo_stays_in_sync():
    res = _run_workers(_ppo_worker, 29613)
    s0, _ = res[0]
    s1, _ = res[1]
    for k in s0:
        torch.testing.assert_close(s0[k], s1[k], rtol=0, atol=0)
