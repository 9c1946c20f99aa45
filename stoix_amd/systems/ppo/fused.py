"""Fused MI355X update path for Anakin PPO (continuous tanh-normal AND
discrete categorical heads on the canonical MLP torsos).

The eager path (ff_ppo.py) runs ~45 kernels per rollout step and ~150 per
minibatch even under hip-graph replay — execution time is dominated by
thousands of 4 us elementwise kernels (profiles/r01_ppo_ant_SUMMARY.md).
This engine replaces it with the hand-written CDNA4 kernels of
stoix_amd/ops/csrc/mlp.hip:

  rollout step:  rollout_step_ant (megakernel: actor fwd + Philox
                 tanh-normal sample + log-prob + critic fwd + quad-lane
                 physics + autoreset + bootstrap critic, ONE launch); or
                 policy_value_step -> env HIP step -> value_forward for
                 non-Ant HIP envs
  minibatch:     ppo_gather -> layer-1 fused linear_silu per net (custom
                 MFMA wins at K=32) -> layer-2 tuned hipBLASLt GemmAndBias
                 per net + ONE stacked silu (library wins at K=256)
                 -> head GEMMs -> ppo_head_loss (per-row losses + analytic
                 head bwd; metrics only on the reported minibatch)
                 -> stacked backward over [2, S, H] buffers: one silu_bwd
                 per layer pair, dH1 as one batched bmm against the
                 adjacent-W2 [2, H, H] view, split-K wgrads into per-chain
                 slabs -> slab_reduce per chain
                 -> ONE bf16 RCCL all-reduce over both chains' shared
                 grad buffer
                 -> fused_adam_bf16 per chain (clip + Adam + bf16 mirror;
                 per-chain lr and global-norm clip match the reference's
                 per-network optax chains)

Parameters: fp32 masters live in ONE flat buffer per network (the module
parameters are repointed to views, so evaluator/checkpointing see updates
for free); a flat bf16 mirror (maintained by the Adam kernel epilogue) is
what the GEMMs and rollout kernels read; gradients are written by the wgrad
GEMMs straight into views of one flat bf16 grad buffer (out=), so the whole
backward produces zero standalone cast/copy kernels.

Eligibility (falls back to the eager path otherwise): CUDA device, MLP torso
[H,H] with SiLU and no LayerNorm, H in {128,256}, NormalAffineTanh head with
ACT<=8 OR Categorical head with ACT<=16 (round 2: Gumbel-max sampling +
log-softmax in policy_value_step_disc, exact-entropy categorical loss +
softmax-jacobian backward in ppo_head_loss_disc), ScalarCriticHead, flat
obs <= 128 dims, no observation normalisation.

Algorithm semantics match ff_ppo.py exactly (same losses, same GAE buffers,
same per-minibatch all-reduce + per-chain clip) with ONE documented
divergence: at the tanh BOUNDARY (|y| >= 1-eps) the kernel's log-prob uses
the clamped-atanh density while the eager distribution uses the
reference's CDF-mass branch (distributions.py log_prob). The fused
eligibility regime (Ant-class, healthy sigma) keeps actions interior where
the two coincide; boundary-saturating tasks (e.g. the swing-up stress
test) run the eager path. Numerics are bf16-GEMM class, verified against the eager fp32 path in tests/test_fused_math.py and
tests/test_fused_gpu.py, and learning curves are re-validated after every
numerics change (tools/learncheck.py; profiles/r01_learning_curves.md).
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.networks.heads import NormalAffineTanhDistributionHead, ScalarCriticHead
from stoix_amd.networks.torso import MLPTorso

Tensor = torch.Tensor


def _mlp_linears(torso: nn.Module) -> Optional[List[nn.Linear]]:
    if not isinstance(torso, MLPTorso):
        return None
    lins = [m for m in torso.net if isinstance(m, nn.Linear)]
    acts = [m for m in torso.net if isinstance(m, nn.SiLU)]
    if len(lins) != 2 or len(acts) != 2 or len(list(torso.net)) != 4:
        return None
    return lins


class _Chain:
    """One flat fp32 master + bf16 mirror + bf16 grad + Adam state.

    ``flat16``/``grad16`` may be preallocated slices of a buffer shared
    with the sibling chain: the engine lays actor and critic out
    back-to-back (actor ends with W2, critic starts with W2) so the two
    W2 blocks are ADJACENT — a zero-copy [2, H, H] view batches the dH1
    backward GEMMs into one bmm — and the whole gradient goes out in ONE
    RCCL all-reduce. Adam/clip state stays per chain (separate lr and
    per-network global-norm clip, matching the reference's per-network
    optax chains)."""

    def __init__(
        self,
        specs: List[Tuple[str, Tuple[int, ...]]],
        device,
        lr: float,
        flat16: Optional[Tensor] = None,
        grad16: Optional[Tensor] = None,
    ):
        self.lr = lr
        sizes = [int(torch.prod(torch.tensor(s)).item()) for _, s in specs]
        total = sum(sizes)
        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        self.flat16 = (
            flat16 if flat16 is not None
            else torch.zeros(total, dtype=torch.bfloat16, device=device)
        )
        self.grad16 = (
            grad16 if grad16 is not None
            else torch.zeros(total, dtype=torch.bfloat16, device=device)
        )
        self.m = torch.zeros(total, dtype=torch.float32, device=device)
        self.v = torch.zeros(total, dtype=torch.float32, device=device)
        self.sqnorm = torch.zeros(1, dtype=torch.float32, device=device)
        self.step_t = torch.zeros(1, dtype=torch.int64, device=device)
        self.views: Dict[str, Tensor] = {}
        self.views16: Dict[str, Tensor] = {}
        self.gviews16: Dict[str, Tensor] = {}
        self.offsets: Dict[str, int] = {}
        off = 0
        for (name, shape), n in zip(specs, sizes):
            self.views[name] = self.flat[off : off + n].view(*shape)
            self.views16[name] = self.flat16[off : off + n].view(*shape)
            self.gviews16[name] = self.grad16[off : off + n].view(*shape)
            self.offsets[name] = off
            off += n
        self.numel = total
        # split-K wgrad slab: 64 fp32 partial-gradient images of the whole
        # chain (one per wave-slice; wgrad.hip WG_SLICES); slab_reduce sums
        # them into grad16
        self.slab = torch.zeros(64, total, dtype=torch.float32, device=device)

    def sync_mirror(self) -> None:
        self.flat16.copy_(self.flat)


class FusedPPOEngine:
    """Drives the fused rollout + update for a PPOLearner. Build with
    :func:`try_build`; returns None when the config isn't eligible."""

    @staticmethod
    def try_build(learner) -> Optional["FusedPPOEngine"]:
        if learner.device.type != "cuda":
            return None
        if type(learner).policy_loss is not _base_policy_loss_func(learner):
            # PPO-penalty / DPO override policy_loss; the fused head kernel
            # implements the clip loss only.
            return None
        actor, critic = learner.actor, learner.critic
        if actor.input_layer is not None or critic.input_layer is not None:
            return None
        a_lins = _mlp_linears(actor.torso)
        c_lins = _mlp_linears(critic.torso)
        if a_lins is None or c_lins is None:
            return None
        head = actor.action_head
        discrete = learner._discrete
        if discrete:
            from stoix_amd.networks.heads import CategoricalHead

            if not isinstance(head, CategoricalHead):
                return None
            ACT = head.linear.out_features
            if ACT > 16:
                return None
        else:
            if not isinstance(head, NormalAffineTanhDistributionHead):
                return None
            ACT = head.loc.out_features
            if ACT > 8:
                return None
        if not isinstance(critic.critic_head, ScalarCriticHead):
            return None
        H = a_lins[0].out_features
        if H not in (128, 256, 512) or a_lins[1].out_features != H:
            return None
        if c_lins[0].out_features != H or c_lins[1].out_features != H:
            return None
        OBS = a_lins[0].in_features
        if OBS > 128:
            return None
        if getattr(learner.env, "_hip", None) is None:
            return None
        from stoix_amd import ops

        if not ops.have_ext():
            return None
        import os

        try:
            return FusedPPOEngine(learner, a_lins, c_lins, H, OBS, ACT)
        except Exception as e:
            if os.environ.get("STOIX_FUSED_STRICT"):
                raise
            import warnings

            warnings.warn(f"fused PPO engine unavailable, eager path: {e!r}")
            return None

    @staticmethod
    def _enable_tunableop() -> None:
        """Load the shipped hipBLASLt TunableOp selections for the fused
        update's GEMM shapes (tuned once on MI355X; wgrad TN at K=32768 is
        ~35% faster than the default pick). Tuning itself stays off."""
        try:
            from pathlib import Path

            tun = torch.cuda.tunable
            if not tun.is_enabled():
                tun.enable(True)
                tun.tuning_enable(False)
            csv = Path(__file__).parent.parent.parent / "ops" / "tunableop_gfx950.csv"
            if csv.exists():
                tun.read_file(str(csv))
        except Exception:
            pass

    def __init__(self, learner, a_lins, c_lins, H: int, OBS: int, ACT: int):
        from stoix_amd import ops

        self.ext = ops.ext(required=True)
        self._enable_tunableop()
        self.learner = learner
        self.device = learner.device
        self.H, self.OBS, self.ACT = H, OBS, ACT
        self.discrete = bool(learner._discrete)
        sysc = learner.sys
        self.clip_eps = float(sysc.clip_eps)
        self.ent_coef = float(sysc.ent_coef)
        self.vf_coef = float(sysc.vf_coef)
        self.max_grad_norm = float(sysc.max_grad_norm)
        head = learner.actor.action_head
        if self.discrete:
            self.min_scale = 0.0
            self.aff_scale = 1.0
            self.aff_shift = 0.0
            self.log_aff_scale = 0.0
        else:
            self.min_scale = float(head.min_scale)
            self.aff_scale = (float(head.maximum) - float(head.minimum)) / 2.0
            self.aff_shift = (float(head.maximum) + float(head.minimum)) / 2.0
            self.log_aff_scale = math.log(self.aff_scale)
        # rank-dependent Philox seed (torch.manual_seed is rank-offset)
        self.seed = int(torch.initial_seed()) % (2**62) + 101
        # first-layer K padded to the MFMA K-step (32): the rollout kernel
        # reads W1 with row stride K1P (mlp.hip load_w_frag), and the
        # update GEMMs use a matching zero-padded obs matrix, so pad
        # columns carry exact zeros through weights, grads and Adam.
        self.K1P = (OBS + 31) & ~31

        dev = self.device
        K1P = self.K1P
        # layout: actor ENDS with W2, critic STARTS with W2 -> adjacent in
        # the shared bf16 buffers (see _Chain docstring)
        a_specs = [
            ("W1", (H, K1P)),
            ("b1", (H,)),
            ("Wh", (16, H)),
            ("bh", (16,)),
            ("b2", (H,)),
            ("W2", (H, H)),
        ]
        c_specs = [
            ("W2", (H, H)),
            ("W1", (H, K1P)),
            ("b1", (H,)),
            ("b2", (H,)),
            ("Wv", (H,)),
            ("bv", (1,)),
        ]
        nA = sum(int(torch.prod(torch.tensor(s)).item()) for _, s in a_specs)
        nC = sum(int(torch.prod(torch.tensor(s)).item()) for _, s in c_specs)
        self.big16 = torch.zeros(nA + nC, dtype=torch.bfloat16, device=dev)
        self.biggrad16 = torch.zeros(nA + nC, dtype=torch.bfloat16, device=dev)
        self.actor_chain = _Chain(
            a_specs, dev, float(sysc.actor_lr),
            flat16=self.big16[:nA], grad16=self.biggrad16[:nA],
        )
        self.critic_chain = _Chain(
            c_specs, dev, float(sysc.critic_lr),
            flat16=self.big16[nA:], grad16=self.biggrad16[nA:],
        )
        # zero-copy [2, H, H] view over [actor W2 | critic W2]
        self.W2pair = self.big16[nA - H * H : nA + H * H].view(2, H, H)
        self._adopt_params(a_lins, c_lins, head, learner.critic.critic_head)
        self.actor_chain.sync_mirror()
        self.critic_chain.sync_mirror()

        # rollout-side buffers
        B = learner.B
        self.empty = torch.zeros(0, device=dev)
        self.draw_policy = torch.zeros(1, dtype=torch.int32, device=dev)
        self.draw_ent = torch.zeros(1, dtype=torch.int32, device=dev)
        self.buf_steptype = torch.zeros(learner.T, B, dtype=torch.uint8, device=dev)
        self.vT = torch.zeros(B, device=dev)  # V(obs_T) for the last step
        self.metrics = torch.zeros(3, dtype=torch.float32, device=dev)
        self.metrics_none = torch.zeros(0, dtype=torch.float32, device=dev)
        self.metric_views = {
            "actor_loss": self.metrics[0],
            "value_loss": self.metrics[1],
            "entropy": self.metrics[2],
        }

        # minibatch workspaces: TWO sets (ping/pong) so the next
        # minibatch's gather can prefetch on a side stream while the
        # current one is in backward (the gather is latency-bound random
        # reads — ~17 us/mb of otherwise dead time)
        S = (learner.T * B) // int(sysc.num_minibatches)
        self.S = S
        z = lambda *s, dtype=torch.bfloat16: torch.zeros(*s, dtype=dtype, device=dev)

        class _Ws:
            def __init__(ws):
                ws.Xmb = z(S, self.K1P)  # K-padded GEMM input (zeros past OBS)
                if self.discrete:
                    ws.act = torch.zeros(S, dtype=torch.long, device=dev)
                else:
                    ws.act = z(S, ACT, dtype=torch.float32)
                ws.logp = z(S, dtype=torch.float32)
                ws.val = z(S, dtype=torch.float32)
                ws.adv = z(S, dtype=torch.float32)
                ws.tgt = z(S, dtype=torch.float32)

        self.ws = [_Ws(), _Ws()]
        # single-set aliases (perf probes / tests reach these)
        self.Xmb = self.ws[0].Xmb
        self.act_mb = self.ws[0].act
        self.logp_mb = self.ws[0].logp
        self.val_mb = self.ws[0].val
        self.adv_mb = self.ws[0].adv
        self.tgt_mb = self.ws[0].tgt
        self.gather_stream = torch.cuda.Stream(dev)
        self.ev_ready = [torch.cuda.Event(), torch.cuda.Event()]
        self.ev_consumed = [torch.cuda.Event(), torch.cuda.Event()]
        # stacked [net, S, H] activation buffers: index 0 = actor, 1 =
        # critic; lets silu_bwd run ONCE over both nets and dH1 run as one
        # batched bmm against W2pair
        self.Z1, self.H1 = z(2, S, H), z(2, S, H)
        self.Z2, self.H2 = z(2, S, H), z(2, S, H)
        self.dH2 = z(2, S, H)
        self.dZ2 = z(2, S, H)
        self.dH1 = z(2, S, H)
        self.dZ1 = z(2, S, H)
        self.heads = z(S, 16)
        self.vpred = z(S, 1)
        self.dhead = z(S, 16)
        self.dv = z(S, 1)
        self.dv16 = z(S, 16)  # col 0 = dv (wgrad A-operand; cols 1-15 zero)

        import os as _os

        # batched dH1 GEMM (one bmm vs two mm); A/B knob for measurement
        self.use_bmm = _os.environ.get("STOIX_FUSED_BMM", "1") != "0"
        # side-stream gather prefetch. Measured SLOWER (28.0 vs 26.8
        # ms/step): the backward kernels already fill all 256 CUs, so the
        # concurrent gather steals bandwidth instead of hiding latency.
        # Kept behind a knob as a documented negative result.
        self.prefetch = _os.environ.get("STOIX_FUSED_PREFETCH", "0") == "1"

        import torch.distributed as dist

        self.world = dist.get_world_size() if dist.is_initialized() else 1

        # observation normalisation (reference ff_ppo.py:90-162): the
        # kernels read STABLE mean/var buffers (stage_obs / ppo_gather
        # nmean/nvar args, plumbed since round 1); the Welford update runs
        # in place after each rollout. Semantics match the reference
        # ordering: act with CURRENT stats, normalise the stored
        # trajectory for the update with the PRE-update stats (gmean/gvar
        # snapshot), THEN update the stats from the raw trajectory.
        self.norm = bool(getattr(learner, "normalize_obs", False))
        if self.norm:
            st = learner.obs_stats
            self.nmean = st.mean  # stable (updated in place by rs.update_)
            self.nvar = (st.std ** 2).clone()
            self.gmean = st.mean.clone()
            self.gvar = self.nvar.clone()

        # the rollout reads the env's stable obs buffer directly; seed it
        # with the current observation (reset happened before attach)
        learner.env._hb["obs"].copy_(learner.cur_obs)

    # ------------------------------------------------------------- params

    def _adopt_params(self, a_lins, c_lins, head, critic_head) -> None:
        """Copy module params into the flat masters and repoint the module
        parameters to views (evaluator / checkpointing then see fused
        updates with no extra copies)."""
        ac, cc = self.actor_chain, self.critic_chain
        ACT, H = self.ACT, self.H
        OBS = self.OBS
        with torch.no_grad():
            ac.views["W1"].zero_()
            cc.views["W1"].zero_()
            ac.views["W1"][:, :OBS].copy_(a_lins[0].weight)
            ac.views["b1"].copy_(a_lins[0].bias)
            ac.views["W2"].copy_(a_lins[1].weight)
            ac.views["b2"].copy_(a_lins[1].bias)
            ac.views["Wh"].zero_()
            ac.views["bh"].zero_()
            if self.discrete:
                # categorical head: logits rows 0..A-1 of the 16-row tile
                ac.views["Wh"][0:ACT].copy_(head.linear.weight)
                ac.views["bh"][0:ACT].copy_(head.linear.bias)
            else:
                ac.views["Wh"][0:ACT].copy_(head.loc.weight)
                ac.views["Wh"][8 : 8 + ACT].copy_(head.scale.weight)
                ac.views["bh"][0:ACT].copy_(head.loc.bias)
                ac.views["bh"][8 : 8 + ACT].copy_(head.scale.bias)
            cc.views["W1"][:, :OBS].copy_(c_lins[0].weight)
            cc.views["b1"].copy_(c_lins[0].bias)
            cc.views["W2"].copy_(c_lins[1].weight)
            cc.views["b2"].copy_(c_lins[1].bias)
            cc.views["Wv"].copy_(critic_head.linear.weight.view(-1))
            cc.views["bv"].copy_(critic_head.linear.bias)
        a_lins[0].weight.data = ac.views["W1"][:, :OBS]
        a_lins[0].bias.data = ac.views["b1"]
        a_lins[1].weight.data = ac.views["W2"]
        a_lins[1].bias.data = ac.views["b2"]
        if self.discrete:
            head.linear.weight.data = ac.views["Wh"][0:ACT]
            head.linear.bias.data = ac.views["bh"][0:ACT]
        else:
            head.loc.weight.data = ac.views["Wh"][0:ACT]
            head.scale.weight.data = ac.views["Wh"][8 : 8 + ACT]
            head.loc.bias.data = ac.views["bh"][0:ACT]
            head.scale.bias.data = ac.views["bh"][8 : 8 + ACT]
        c_lins[0].weight.data = cc.views["W1"][:, :OBS]
        c_lins[0].bias.data = cc.views["b1"]
        c_lins[1].weight.data = cc.views["W2"]
        c_lins[1].bias.data = cc.views["b2"]
        critic_head.linear.weight.data = cc.views["Wv"].view(1, H)
        critic_head.linear.bias.data = cc.views["bv"]

    def refresh_masters(self) -> None:
        """After external writes to the module params (checkpoint restore),
        re-sync the bf16 mirrors."""
        self.actor_chain.sync_mirror()
        self.critic_chain.sync_mirror()

    # ------------------------------------------------------------ rollout

    def rollout(self) -> None:
        L = self.learner
        env = L.env
        hb = env._hb
        ac, cc = self.actor_chain, self.critic_chain
        a16, c16 = ac.views16, cc.views16
        ext = self.ext
        nmean = self.nmean if self.norm else self.empty
        nvar = self.nvar if self.norm else self.empty
        if getattr(env, "HIP_KERNEL", "") == "ant_step" and not self.norm:
            # megakernel path: ONE launch per rollout step (policy + Ant
            # physics + bootstrap critic; mlp.hip rollout_step_ant_kernel)
            for t in range(L.T):
                ext.rollout_step_ant(
                    hb["obs"], env._state["s"], env._step_count,
                    env._ep_return, env._ep_length, env._last_ep_return,
                    env._last_ep_length,
                    a16["W1"], ac.views["b1"], a16["W2"], ac.views["b2"],
                    a16["Wh"], ac.views["bh"],
                    c16["W1"], cc.views["b1"], c16["W2"], cc.views["b2"],
                    c16["Wv"], cc.views["bv"],
                    L.buf_obs[t], L.buf_action[t], L.buf_log_prob[t],
                    L.buf_value[t], L.buf_bootstrap[t], L.buf_reward[t],
                    L.buf_discount[t], self.buf_steptype[t],
                    env.max_episode_steps, self.min_scale, self.aff_scale,
                    self.aff_shift, self.log_aff_scale, self.seed,
                    env._hip_seed, self.draw_policy, hb["draw"], t,
                )
            # bootstrap fill: the megakernel wrote V(next_obs) for DONE
            # rows only; for every non-done row V(next_obs) == the next
            # step's stored value (same weights, identical observation) —
            # one shifted masked copy + a single B-row critic pass for the
            # final step's non-done rows
            ext.value_forward(
                hb["obs"], c16["W1"], cc.views["b1"], c16["W2"],
                cc.views["b2"], c16["Wv"], cc.views["bv"], self.vT,
                self.empty, self.empty,
            )
            done = self.buf_steptype >= 2  # TERMINATED(2) | TRUNCATED(3)
            torch.where(done[:-1], L.buf_bootstrap[:-1], L.buf_value[1:],
                        out=L.buf_bootstrap[:-1])
            torch.where(done[-1], L.buf_bootstrap[-1], self.vT,
                        out=L.buf_bootstrap[-1])
            ext.bump_add(self.draw_policy, L.T)
            ext.bump_add(hb["draw"], L.T)
            L.buf_truncated.copy_(self.buf_steptype == 3)
            env._done_count += done.sum()
            return
        for t in range(L.T):
            if self.discrete:
                ext.policy_value_step_disc(
                    hb["obs"],
                    a16["W1"], ac.views["b1"], a16["W2"], ac.views["b2"],
                    a16["Wh"], ac.views["bh"],
                    c16["W1"], cc.views["b1"], c16["W2"], cc.views["b2"],
                    c16["Wv"], cc.views["bv"],
                    L.buf_obs[t], L.buf_action[t], L.buf_log_prob[t],
                    L.buf_value[t], nmean, nvar,
                    self.ACT, 0, self.seed, self.draw_policy, t, 0,
                )
            else:
                ext.policy_value_step(
                    hb["obs"],
                    a16["W1"], ac.views["b1"], a16["W2"], ac.views["b2"],
                    a16["Wh"], ac.views["bh"],
                    c16["W1"], cc.views["b1"], c16["W2"], cc.views["b2"],
                    c16["Wv"], cc.views["bv"],
                    L.buf_obs[t], L.buf_action[t], L.buf_log_prob[t], L.buf_value[t],
                    nmean, nvar,
                    self.min_scale, self.aff_scale, self.aff_shift,
                    self.log_aff_scale, 0, self.seed, self.draw_policy, t, 0,
                )
            env.hip_step_into(
                L.buf_action[t],
                reward_out=L.buf_reward[t],
                discount_out=L.buf_discount[t],
                steptype_out=self.buf_steptype[t],
                draw_offset=t,
                do_bump=False,
            )
            ext.value_forward(
                hb["next_obs"],
                c16["W1"], cc.views["b1"], c16["W2"], cc.views["b2"],
                c16["Wv"], cc.views["bv"],
                L.buf_bootstrap[t], nmean, nvar,
            )
        # one counter bump per rollout (the per-step draws used frozen
        # offsets t baked into the graph nodes)
        ext.bump_add(self.draw_policy, L.T)
        ext.bump_add(hb["draw"], L.T)
        # truncation flags for GAE (StepType.TRUNCATED == 3)
        L.buf_truncated.copy_(self.buf_steptype == 3)
        if self.norm:
            self._update_obs_stats()

    def _update_obs_stats(self) -> None:
        """Snapshot the PRE-update stats for the epoch gathers, then run
        the in-place Welford update over the raw stored trajectory
        (reference ff_ppo.py:150-162 ordering)."""
        from stoix_amd.ops import running_statistics as rs

        L = self.learner
        self.gmean.copy_(self.nmean)
        self.gvar.copy_(self.nvar)
        rs.update_(L.obs_stats, L.buf_obs, all_reduce=self.world > 1)
        torch.pow(L.obs_stats.std, 2, out=self.nvar)

    # ------------------------------------------------------------- update

    def epoch(self) -> Dict[str, Tensor]:
        L = self.learner
        ext = self.ext
        TB = L.T * L.B
        n_mb = int(L.sys.num_minibatches)
        S = self.S
        OBSd = self.OBS
        flat_obs = L.buf_obs.view(TB, OBSd)
        flat_action = (
            L.buf_action.view(TB) if self.discrete
            else L.buf_action.view(TB, self.ACT)
        )
        flat_logp = L.buf_log_prob.view(TB)
        flat_value = L.buf_value.view(TB)
        flat_adv = L.buf_adv.view(TB)
        flat_tgt = L.buf_targets.view(TB)
        ac, cc = self.actor_chain, self.critic_chain
        a16, c16 = ac.views16, cc.views16
        ag, cg = ac.gviews16, cc.gviews16

        import torch.distributed as dist

        gather_fn = ext.ppo_gather_disc if self.discrete else ext.ppo_gather
        gmean = self.gmean if self.norm else self.empty
        gvar = self.gvar if self.norm else self.empty

        def gather_into(mb: int, w) -> None:
            idx = L.perm_buf[mb * S : (mb + 1) * S]
            gather_fn(
                idx, flat_obs, flat_action, flat_logp, flat_value, flat_adv,
                flat_tgt, w.Xmb, w.act, w.logp, w.val, w.adv, w.tgt,
                gmean, gvar,
            )

        # gather prefetch: minibatch mb+1's gather runs on a side stream
        # while mb's backward is on the main stream; double-buffered
        # workspaces + events make the dependency explicit (capture-legal:
        # record/wait become hipGraph edges)
        gs = self.gather_stream
        cur = torch.cuda.current_stream(self.device)
        prefetch = self.prefetch
        if prefetch:
            gs.wait_stream(cur)
            with torch.cuda.stream(gs):
                gather_into(0, self.ws[0])
                self.ev_ready[0].record(gs)

        for mb in range(n_mb):
            p = mb & 1 if prefetch else 0
            w = self.ws[p]
            if prefetch:
                cur.wait_event(self.ev_ready[p])
            else:
                gather_into(mb, w)
            # ---- forward. Layer 1 (K=32): the custom fused Linear+SiLU
            # MFMA kernel wins (14.4 us vs 11.2 GEMM + silu pass). Layer 2
            # (K=256): hipBLASLt's tuned GemmAndBias (14.4 us) beats the
            # custom kernel (24.6 us), so run both nets' GEMMs then ONE
            # silu over the stacked [2, S, H] preacts.
            ext.linear_silu(w.Xmb, a16["W1"], ac.views["b1"], self.Z1[0], self.H1[0], 1)
            ext.linear_silu(w.Xmb, c16["W1"], cc.views["b1"], self.Z1[1], self.H1[1], 1)
            torch.addmm(a16["b2"], self.H1[0], a16["W2"].t(), out=self.Z2[0])
            torch.addmm(c16["b2"], self.H1[1], c16["W2"].t(), out=self.Z2[1])
            ext.silu_fwd(self.Z2, self.H2)
            # ---- heads as GEMMs (hipBLASLt), then the fused per-row
            # loss + analytic head-backward kernel, then dH2 as GEMMs
            torch.addmm(a16["bh"], self.H2[0], a16["Wh"].t(), out=self.heads)
            torch.addmm(c16["bv"], self.H2[1], c16["Wv"].view(self.H, 1),
                        out=self.vpred)
            # metrics only on the LAST minibatch (the one epoch() reports):
            # 512 waves of atomics into 3 words serialise across XCDs
            last = mb == n_mb - 1
            if last:
                self.metrics.zero_()
            if self.discrete:
                ext.ppo_head_loss_disc(
                    self.heads, self.vpred.view(-1), w.act, w.logp,
                    w.val, w.adv, w.tgt, self.dhead, self.dv,
                    self.dv16, self.metrics if last else self.metrics_none,
                    self.ACT, self.clip_eps, self.ent_coef, self.vf_coef,
                )
            else:
                ext.ppo_head_loss(
                    self.heads, self.vpred.view(-1), w.act, w.logp,
                    w.val, w.adv, w.tgt, self.dhead, self.dv,
                    self.dv16, self.metrics if last else self.metrics_none,
                    self.clip_eps, self.ent_coef,
                    self.vf_coef, self.min_scale, self.aff_scale, self.aff_shift,
                    self.log_aff_scale, self.seed, self.draw_ent, mb, 0,
                )
            torch.mm(self.dhead, a16["Wh"], out=self.dH2[0])
            torch.mm(self.dv, c16["Wv"].view(1, self.H), out=self.dH2[1])
            # ---- backward: wgrads + bias colsums via the split-K MFMA
            # wgrad kernel (wgrad.hip; hipBLASLt NT at K=32768 is ~3x
            # slower), partial sums land in the chain slab, one
            # slab_reduce per chain folds them into the flat bf16 grads.
            # silu_bwd runs once over the stacked [2, S, H] buffers and the
            # two dH1 GEMMs go as one bmm against the adjacent-W2 view.
            ao, co = ac.offsets, cc.offsets
            ext.wgrad(self.dhead, self.H2[0], ac.slab, ao["Wh"], ao["bh"], 16)
            ext.wgrad(self.dv16, self.H2[1], cc.slab, co["Wv"], co["bv"], 1)
            ext.silu_bwd(self.dH2, self.Z2, self.dZ2)
            ext.wgrad(self.dZ2[0], self.H1[0], ac.slab, ao["W2"], ao["b2"], self.H)
            ext.wgrad(self.dZ2[1], self.H1[1], cc.slab, co["W2"], co["b2"], self.H)
            if self.use_bmm:
                torch.bmm(self.dZ2, self.W2pair, out=self.dH1)
            else:
                torch.mm(self.dZ2[0], a16["W2"], out=self.dH1[0])
                torch.mm(self.dZ2[1], c16["W2"], out=self.dH1[1])
            ext.silu_bwd(self.dH1, self.Z1, self.dZ1)
            ext.wgrad(self.dZ1[0], w.Xmb, ac.slab, ao["W1"], ao["b1"], self.H)
            ext.wgrad(self.dZ1[1], w.Xmb, cc.slab, co["W1"], co["b1"], self.H)
            if prefetch:
                # all reads of workspace set p are done (the W1 wgrads were
                # the last); let the side stream refill it for mb+2, and
                # kick off mb+1's gather now
                self.ev_consumed[p].record(cur)
                if mb + 1 < n_mb:
                    q = (mb + 1) & 1
                    gs.wait_event(self.ev_consumed[q])
                    with torch.cuda.stream(gs):
                        gather_into(mb + 1, self.ws[q])
                        self.ev_ready[q].record(gs)
            ext.slab_reduce(ac.slab, ac.grad16, ac.sqnorm, ac.step_t)
            ext.slab_reduce(cc.slab, cc.grad16, cc.sqnorm, cc.step_t)
            # ---- ONE all-reduce over both chains' shared grad buffer,
            # then fused clip/Adam per chain (+ bf16 mirror refresh)
            if self.world > 1:
                dist.all_reduce(self.biggrad16)
            gscale = 1.0 / float(self.world)
            ext.fused_adam_bf16(
                ac.flat, ac.grad16, ac.m, ac.v, ac.sqnorm, ac.step_t,
                ac.flat16, ac.lr, 0.9, 0.999, 1e-5, self.max_grad_norm,
                gscale, 0,
            )
            ext.fused_adam_bf16(
                cc.flat, cc.grad16, cc.m, cc.v, cc.sqnorm, cc.step_t,
                cc.flat16, cc.lr, 0.9, 0.999, 1e-5, self.max_grad_norm,
                gscale, 0,
            )
        if prefetch:
            cur.wait_stream(gs)
        ext.bump_add(self.draw_ent, n_mb)
        return {k: v for k, v in self.metric_views.items()}


def _base_policy_loss_func(learner):
    """The PPOLearner base policy_loss function (unbound)."""
    from stoix_amd.systems.ppo import ff_ppo

    return ff_ppo.PPOLearner.policy_loss
