"""End-to-end CI gate for the driver bench contract.

Runs bench.py exactly the way the round-end driver does — under
``python -m torch.distributed.run --nnodes=1 --nproc-per-node 2`` with
``--master-addr 127.0.0.1`` — on CPU (gloo world=2) and asserts ONE valid
JSON line with the whole-job aggregate. This pins the torchrun rendezvous,
per-rank env plumbing, warmup collective, barrier+MAX timing and JSON
emission before they ever meet 8 GPUs.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(extra, nproc=2, timeout=420):
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={nproc}",
        "--master-addr=127.0.0.1",
        "--master-port=29551",
        os.path.join(REPO, "bench.py"),
        f"--gpus={nproc}",
    ] + extra
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        cmd, capture_output=True, text=True, timeout=timeout, cwd=REPO, env=env
    )
    assert out.returncode == 0, f"bench failed:\n{out.stdout}\n{out.stderr}"
    lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {out.stdout!r}"
    return json.loads(lines[0])


@pytest.mark.slow
def test_bench_world2_gloo_cpu():
    rec = _run_bench(
        ["--steps=2", "--warmup=1", "--num-envs=8", "--rollout-length=8"]
    )
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 2
    assert rec["unit"] == "env_steps/s"
    assert rec["value"] > 0
    # whole-job aggregate: 2 steps * T=8 * 8 envs * 2 ranks env-steps total
    assert abs(rec["value"] * rec["ms_per_step"] / 1000.0 * rec["steps"] - 2 * 8 * 8 * 2) < 1e-3
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["scaling"] == "weak"


@pytest.mark.slow
def test_bench_world1_single_process():
    out = subprocess.run(
        [
            sys.executable,
            os.path.join(REPO, "bench.py"),
            "--steps=2",
            "--warmup=1",
            "--num-envs=4",
            "--rollout-length=8",
        ],
        capture_output=True,
        text=True,
        timeout=420,
        cwd=REPO,
    )
    assert out.returncode == 0, out.stderr
    rec = json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert rec["n_gpus"] == 1 and rec["value"] > 0


@pytest.mark.slow
def test_bench_world2_sac_and_rainbow_builders():
    """The non-flagship bench builders also honour the torchrun world=2
    contract (device-pinned learners, barrier+MAX timing, one JSON line)."""
    rec = _run_bench(
        ["--config=sac", "--steps=2", "--warmup=1", "--num-envs=8"],
        timeout=600,
    )
    assert rec["n_gpus"] == 2 and rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp2"
    rec = _run_bench(
        ["--config=rainbow", "--steps=2", "--warmup=1", "--num-envs=8"],
        timeout=600,
    )
    assert rec["n_gpus"] == 2 and rec["value"] > 0
