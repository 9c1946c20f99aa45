"""SLURM launcher (parity: /root/reference/stoix/slurm_launcher.py, which
uses submitit; submitit is not installable offline, so this generates and
submits standard sbatch scripts directly).

Launches one task per node with `torch.distributed.run` spawning one rank
per GPU (the MI355X execution shape: 8 ranks over RCCL/xGMI per node).

Usage:
    python -m stoix_amd.slurm_launcher \
        --entry stoix_amd.systems.ppo.ff_ppo \
        --partition amd --gpus-per-node 8 --nodes 1 \
        -- env=brax/ant arch.seed=0,1,2

Comma-separated override values fan out into a job array (one run per
combination, like the reference's submitit sweeps).
"""
from __future__ import annotations

import argparse
import itertools
import shutil
import subprocess
import sys
from pathlib import Path
from typing import List


SBATCH_TEMPLATE = """#!/bin/bash
#SBATCH --job-name={name}
#SBATCH --partition={partition}
#SBATCH --nodes={nodes}
#SBATCH --gpus-per-node={gpus}
#SBATCH --cpus-per-task={cpus}
#SBATCH --time={time}
#SBATCH --output={logdir}/%x_%A_%a.out
{array_line}
set -euo pipefail
export HSA_ENABLE_IPC_MODE_LEGACY=0
OVERRIDES=({override_lines})
RUN_OVERRIDES=${{OVERRIDES[${{SLURM_ARRAY_TASK_ID:-0}}]}}
exec python -m torch.distributed.run \\
    --nnodes={nodes} --nproc-per-node={gpus} \\
    --master-addr "${{SLURM_LAUNCH_NODE_IPADDR:-127.0.0.1}}" \\
    --master-port {port} \\
    -m {entry} ${{RUN_OVERRIDES}}
"""


def expand_overrides(overrides: List[str]) -> List[List[str]]:
    """a=1,2 b=x -> [[a=1, b=x], [a=2, b=x]] (submitit-style sweep fan-out)."""
    keys, values = [], []
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override '{ov}' is not key=value")
        k, v = ov.split("=", 1)
        keys.append(k)
        values.append(v.split(","))
    runs = []
    for combo in itertools.product(*values) if values else [()]:
        runs.append([f"{k}={v}" for k, v in zip(keys, combo)])
    return runs


def build_script(args, runs: List[List[str]]) -> str:
    array_line = f"#SBATCH --array=0-{len(runs) - 1}" if len(runs) > 1 else ""
    override_lines = " ".join('"' + " ".join(r) + '"' for r in runs)
    return SBATCH_TEMPLATE.format(
        name=args.name,
        partition=args.partition,
        nodes=args.nodes,
        gpus=args.gpus_per_node,
        cpus=args.cpus_per_task,
        time=args.time,
        logdir=args.logdir,
        array_line=array_line,
        override_lines=override_lines,
        port=args.port,
        entry=args.entry,
    )


def main(argv=None) -> int:
    argv = sys.argv[1:] if argv is None else argv
    if "--" in argv:
        split = argv.index("--")
        argv, overrides = argv[:split], argv[split + 1 :]
    else:
        overrides = []
    p = argparse.ArgumentParser()
    p.add_argument("--entry", required=True, help="python module with a hydra_entry_point")
    p.add_argument("--name", default="stoix_amd")
    p.add_argument("--partition", default="amd")
    p.add_argument("--nodes", type=int, default=1)
    p.add_argument("--gpus-per-node", type=int, default=8)
    p.add_argument("--cpus-per-task", type=int, default=32)
    p.add_argument("--time", default="24:00:00")
    p.add_argument("--port", type=int, default=29517)
    p.add_argument("--logdir", default="slurm_logs")
    p.add_argument("--dry-run", action="store_true", help="write the script, do not sbatch")
    args = p.parse_args(argv)

    runs = expand_overrides(overrides)
    script = build_script(args, runs)
    Path(args.logdir).mkdir(parents=True, exist_ok=True)
    path = Path(args.logdir) / f"{args.name}.sbatch"
    path.write_text(script)
    print(f"wrote {path} ({len(runs)} run(s))")
    if args.dry_run or shutil.which("sbatch") is None:
        if not args.dry_run:
            print("sbatch not found on PATH; use --dry-run semantics (script written)")
        return 0
    out = subprocess.run(["sbatch", str(path)], capture_output=True, text=True)
    print(out.stdout.strip() or out.stderr.strip())
    return out.returncode


if __name__ == "__main__":
    raise SystemExit(main())
