"""hip-graph capture of the Anakin update step.

The Anakin inner loop is launch-latency-bound: a rollout of T=128 steps with
tiny MLPs issues thousands of kernels whose boundaries (~1.2-1.5 us each,
MI355X_MICROARCH 'boundary') and host launch overhead (~3.3-3.8 us eager)
dominate. Capture replaces thousands of host launches per update with a
handful of graph replays (~10-16 us each) — the MI355X equivalent of the
reference's one-XLA-program design.

Two graphs per learner:
  * rollout graph — T env steps (fused HIP kernels, device-side RNG
    counters) + actor/critic forwards + GAE, writing stable buffers;
  * epoch graph — num_minibatches x (forward, losses, backward, RCCL
    all-reduce, clip, Adam) reading a static permutation buffer.
The minibatch permutation itself is NOT capture-legal at scale (randperm /
sort do host work), so it is refreshed eagerly into ``perm_buf`` between
replays — one tensor write per epoch.

Capture preconditions handled here: the env must have the fused HIP step
path; the learner's RNG falls back to the default (graph-aware) CUDA
generator; the gradient all-reduce runs inline on the capture stream (RCCL
supports capture); Adam is in capturable mode; host-side metric reads move
outside the graphs.
"""
from __future__ import annotations

import os
from typing import Dict

import torch


def _world_size() -> int:
    import torch.distributed as dist

    return dist.get_world_size() if dist.is_initialized() else 1


def _warm_collectives(device: torch.device) -> None:
    """Run one eager all-reduce before any capture: the RCCL communicator
    (ring setup over xGMI) must be created OUTSIDE graph capture — the
    first collective builds it, and building it inside a capture fails.
    Also doubles as a fail-fast connectivity check at bench start."""
    import torch.distributed as dist

    if dist.is_initialized() and dist.get_world_size() > 1:
        t = torch.ones(1, device=device)
        dist.all_reduce(t)
        if device.type == "cuda":
            # warm the bf16 path too (the fused engine all-reduces bf16)
            tb = torch.ones(8, dtype=torch.bfloat16, device=device)
            dist.all_reduce(tb)
            torch.cuda.synchronize(device)


def try_enable_graphs(learner) -> bool:
    """Capture learner's rollout/epoch phases into hip graphs and
    monkey-patch update_step to replay them. Returns True on success.

    Knobs (multi-GPU first-try safety):
      STOIX_NO_GRAPH=1             — skip capture entirely (pure eager).
      STOIX_FUSED_EAGER_ALLREDUCE=1 — at world>1, capture only the rollout
        graph and run the epoch phase eagerly, so the per-minibatch RCCL
        all-reduce is an ordinary eager call instead of a graph node.
    """
    device = learner.device
    if device.type != "cuda":
        return False
    if os.environ.get("STOIX_NO_GRAPH"):
        return False
    if getattr(learner, "lr_decay", None) is not None:
        # a python-float lr would be frozen into the captured optimiser
        # step; the linear-decay path stays eager (reference parity:
        # utils/training.py decay is opt-in and off in every shipped config)
        return False
    env = learner.env
    if getattr(env, "_hip", None) is None and not getattr(env, "capture_safe", False):
        raise RuntimeError(
            "env has no HIP step kernel and no capture_safe torch step; "
            "graph capture needs one"
        )

    learner.prepare_for_graph_capture()
    _warm_collectives(device)

    # eager warmup on a side stream (rocBLAS/hipBLASLt workspaces, autotune)
    side = torch.cuda.Stream(device)
    side.wait_stream(torch.cuda.current_stream(device))
    with torch.cuda.stream(side):
        for _ in range(2):
            learner.rollout_phase()
            learner._new_perm()
            learner.epoch_phase()
    torch.cuda.current_stream(device).wait_stream(side)
    torch.cuda.synchronize(device)

    g_rollout = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g_rollout):
        learner.rollout_phase()

    epochs = int(learner.sys.epochs)
    eager_epoch = (
        _world_size() > 1 and os.environ.get("STOIX_FUSED_EAGER_ALLREDUCE") == "1"
    )
    if eager_epoch:

        def update_step() -> Dict[str, torch.Tensor]:
            g_rollout.replay()
            for _ in range(epochs):
                learner._new_perm()
                metrics = learner.epoch_phase()
            learner.after_graph_replay()
            return metrics

        learner._graphs = (g_rollout,)
        learner.update_step = update_step
        return True

    learner._new_perm()
    g_epoch = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g_epoch):
        static_metrics = learner.epoch_phase()

    def update_step() -> Dict[str, torch.Tensor]:
        g_rollout.replay()
        for _ in range(epochs):
            learner._new_perm()
            g_epoch.replay()
        learner.after_graph_replay()
        return static_metrics

    learner._graphs = (g_rollout, g_epoch)
    learner.update_step = update_step
    return True


def try_enable_update_graph(learner) -> bool:
    """Capture an off-policy learner's whole update_step (rollout into the
    device-resident replay buffer + epochs of sample/loss/backward/Adam/
    polyak) into ONE hip graph. Requires: HIP env (no host-side autoreset
    branch), device-cursor ItemBuffer, capturable Adam. The learner opts in
    via ``graph_capturable`` and provides prepare_for_graph_capture /
    after_graph_replay hooks (metrics read eagerly from the env's latched
    buffers after each replay)."""
    device = learner.device
    if device.type != "cuda":
        return False
    if os.environ.get("STOIX_NO_GRAPH"):
        return False
    if _world_size() > 1 and os.environ.get("STOIX_FUSED_EAGER_ALLREDUCE") == "1":
        # whole-update capture would put the RCCL all-reduce inside the
        # graph; the knob demands eager collectives, so skip capture
        return False
    if getattr(learner, "lr_decay", None) is not None:
        return False
    env = learner.env
    if getattr(env, "_hip", None) is None and not getattr(env, "capture_safe", False):
        raise RuntimeError(
            "update-graph capture needs a HIP env step kernel or a "
            "capture_safe torch env step"
        )
    if hasattr(env, "prepare_for_graph_capture"):
        env.prepare_for_graph_capture()
    learner.prepare_for_graph_capture()
    _warm_collectives(device)

    side = torch.cuda.Stream(device)
    side.wait_stream(torch.cuda.current_stream(device))
    with torch.cuda.stream(side):
        for _ in range(2):
            learner.update_step()
    torch.cuda.current_stream(device).wait_stream(side)
    torch.cuda.synchronize(device)

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        static_metrics = learner.update_step()

    def update_step() -> Dict[str, torch.Tensor]:
        g.replay()
        learner.after_graph_replay()
        return static_metrics

    learner._graphs = (g,)
    learner.update_step = update_step
    return True
