"""hip-graph capture of the Anakin update step.

The Anakin inner loop is launch-latency-bound: a rollout of T=128 steps with
tiny MLPs issues thousands of kernels whose boundaries (~1.2-1.5 us each,
MI355X_MICROARCH 'boundary') and host launch overhead (~3.3-3.8 us eager)
dominate. Capturing the WHOLE update step (rollout + GAE + epochs of
minibatch updates) into one hipGraph replays it with a single ~10-16 us
host call — the MI355X equivalent of the reference's one-XLA-program design
(SURVEY.md §3.1).

Capture preconditions handled here:
  * env must have the fused HIP step path (device-side RNG counter) — the
    torch env path has host-dependent control flow and cannot be captured;
  * the learner's torch RNG falls back to the default CUDA generator
    (graph-aware) instead of a user Generator;
  * gradient all-reduce runs on the capture stream (RCCL supports graph
    capture); Adam runs in capturable mode;
  * episode-metric extraction (host reads) moves outside the graph.
"""
from __future__ import annotations

from typing import Dict

import torch


def try_enable_graphs(learner) -> bool:
    """Capture learner.update_step into a hip graph; monkey-patches
    learner.update_step to replay it. Returns True on success."""
    device = learner.device
    if device.type != "cuda":
        return False
    if getattr(learner.env, "_hip", None) is None:
        raise RuntimeError("env has no HIP step kernel; graph capture needs it")

    learner.prepare_for_graph_capture()

    # eager warmup on a side stream (rocBLAS/hipBLASLt workspaces, autotuning)
    side = torch.cuda.Stream(device)
    side.wait_stream(torch.cuda.current_stream(device))
    with torch.cuda.stream(side):
        for _ in range(2):
            learner.graph_body()
    torch.cuda.current_stream(device).wait_stream(side)
    torch.cuda.synchronize(device)

    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        static_metrics = learner.graph_body()

    def update_step() -> Dict[str, torch.Tensor]:
        graph.replay()
        learner.after_graph_replay()
        return static_metrics

    learner._graph = graph
    learner.update_step = update_step
    return True
