"""Per-phase and per-kernel timing probe for the fused PPO path.

Run on a GPU box:  python tools/perf_probe.py
Prints: fused-engine attach status, rollout vs epoch wall times (graphed and
eager), and microbenchmarks of the individual fused kernels.
"""
from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    os.environ["STOIX_FUSED_STRICT"] = "1"
    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        [
            "env=brax/ant",
            "arch.total_num_envs=4096",
            "arch.total_timesteps=null",
            "arch.num_updates=100",
            "arch.num_evaluation=1",
            "system.rollout_length=128",
            "system.num_minibatches=16",
            "system.epochs=4",
            "system.compute_dtype=bf16",
            "logger.loggers=[]",
        ],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    env = environments.make_single(cfg, 4096, device, seed=0)
    learner = PPOLearner(cfg, env, device)
    print("fused attached:", learner.fused is not None)

    t_roll = timeit(learner.rollout_phase, iters=10)
    print(f"rollout_phase (incl GAE): {t_roll:8.2f} ms")
    learner._new_perm()
    t_epoch = timeit(learner.epoch_phase, iters=10)
    print(f"epoch_phase:              {t_epoch:8.2f} ms")
    e = int(cfg.system.epochs)
    print(f"=> step estimate: {t_roll + e * t_epoch:8.2f} ms "
          f"({t_roll:.1f} + {e}x{t_epoch:.1f})")

    if learner.fused is None:
        return
    F = learner.fused
    ext = F.ext
    L = learner
    hb = env._hb
    ac, cc = F.actor_chain, F.critic_chain
    a16, c16 = ac.views16, cc.views16

    def k_policy():
        ext.policy_value_step(
            hb["obs"], a16["W1"], ac.views["b1"], a16["W2"], ac.views["b2"],
            a16["Wh"], ac.views["bh"], c16["W1"], cc.views["b1"], c16["W2"],
            cc.views["b2"], c16["Wv"], cc.views["bv"], L.buf_obs[0],
            L.buf_action[0], L.buf_log_prob[0], L.buf_value[0], F.empty,
            F.empty, F.min_scale, F.aff_scale, F.aff_shift, F.log_aff_scale,
            0, F.seed, F.draw_policy, 0, 1)

    def k_value():
        ext.value_forward(hb["next_obs"], c16["W1"], cc.views["b1"],
                          c16["W2"], cc.views["b2"], c16["Wv"], cc.views["bv"],
                          L.buf_bootstrap[0], F.empty, F.empty)

    def k_env():
        env.hip_step_into(L.buf_action[0], L.buf_reward[0], L.buf_discount[0],
                          F.buf_steptype[0], 0, True)

    print(f"policy_value_step:  {timeit(k_policy, 50)*1e3:8.1f} us")
    print(f"value_forward:      {timeit(k_value, 50)*1e3:8.1f} us")
    print(f"env hip step:       {timeit(k_env, 50)*1e3:8.1f} us")

    # update-phase pieces
    S = F.S
    TB = L.T * L.B
    idx = torch.arange(S, device=device)
    flat_obs = L.buf_obs.view(TB, F.OBS)
    flat_action = L.buf_action.view(TB, F.ACT)
    flat_logp = L.buf_log_prob.view(TB)
    flat_value = L.buf_value.view(TB)
    flat_adv = L.buf_adv.view(TB)
    flat_tgt = L.buf_targets.view(TB)

    def k_gather():
        ext.ppo_gather(idx, flat_obs, flat_action, flat_logp, flat_value,
                       flat_adv, flat_tgt, F.Xmb, F.act_mb, F.logp_mb,
                       F.val_mb, F.adv_mb, F.tgt_mb, F.empty, F.empty)

    def k_gemm_fwd():
        torch.addmm(a16["b1"], F.Xmb, a16["W1"].t(), out=F.Z1[0])

    def k_gemm_fwd2():
        torch.addmm(a16["b2"], F.H1[0], a16["W2"].t(), out=F.Z2[0])

    def k_silu():
        ext.silu_fwd(F.Z1[0], F.H1[0])

    def k_linsilu():
        ext.linear_silu(F.H1[0], a16["W2"], ac.views["b2"], F.Z2[0], F.H2[0], 1)

    def k_headgemm():
        torch.addmm(a16["bh"], F.H2[0], a16["Wh"].t(), out=F.heads)

    def k_head():
        F.metrics.zero_()
        ext.ppo_head_loss(
            F.heads, F.vpred.view(-1), F.act_mb, F.logp_mb, F.val_mb,
            F.adv_mb, F.tgt_mb, F.dhead, F.dv, F.dv16, F.metrics, F.clip_eps,
            F.ent_coef, F.vf_coef, F.min_scale, F.aff_scale, F.aff_shift,
            F.log_aff_scale, F.seed, F.draw_ent, 0, 1)

    def k_wgrad():
        torch.mm(F.dZ2[0].t(), F.H1[0], out=ac.gviews16["W2"])

    def k_wgrad_custom():
        ext.wgrad(F.dZ2[0], F.H1[0], ac.slab, ac.offsets["W2"], ac.offsets["b2"], F.H)

    def k_slab_reduce():
        ext.slab_reduce(ac.slab, ac.grad16, ac.sqnorm, ac.step_t)

    def k_dgrad():
        torch.mm(F.dZ2[0], a16["W2"], out=F.dH1[0])

    def k_dgrad_bmm():
        torch.bmm(F.dZ2, F.W2pair, out=F.dH1)

    def k_silu_bwd_stacked():
        ext.silu_bwd(F.dH2, F.Z2, F.dZ2)

    def k_bsum():
        torch.sum(F.dZ2[0], 0, out=ac.gviews16["b2"])

    def k_adam():
        ext.fused_adam_bf16(ac.flat, ac.grad16, ac.m, ac.v, ac.sqnorm,
                            ac.step_t, ac.flat16, ac.lr, 0.9, 0.999, 1e-5,
                            F.max_grad_norm, 1.0, 1)

    print(f"ppo_gather:         {timeit(k_gather, 50)*1e3:8.1f} us")
    print(f"gemm fwd L1 (pad):  {timeit(k_gemm_fwd, 50)*1e3:8.1f} us")
    print(f"gemm fwd L2:        {timeit(k_gemm_fwd2, 50)*1e3:8.1f} us")
    print(f"silu_fwd:           {timeit(k_silu, 50)*1e3:8.1f} us")
    print(f"linear_silu fused:  {timeit(k_linsilu, 50)*1e3:8.1f} us")
    print(f"head gemm:          {timeit(k_headgemm, 50)*1e3:8.1f} us")
    print(f"ppo_head_loss:      {timeit(k_head, 50)*1e3:8.1f} us")
    print(f"wgrad mm:           {timeit(k_wgrad, 50)*1e3:8.1f} us")
    print(f"wgrad custom:       {timeit(k_wgrad_custom, 50)*1e3:8.1f} us")
    print(f"slab_reduce:        {timeit(k_slab_reduce, 50)*1e3:8.1f} us")
    print(f"dgrad mm:           {timeit(k_dgrad, 50)*1e3:8.1f} us")
    print(f"dgrad bmm stacked:  {timeit(k_dgrad_bmm, 50)*1e3:8.1f} us")
    print(f"silu_bwd stacked:   {timeit(k_silu_bwd_stacked, 50)*1e3:8.1f} us")
    print(f"bias colsum:        {timeit(k_bsum, 50)*1e3:8.1f} us")
    print(f"fused_adam_bf16:    {timeit(k_adam, 50)*1e3:8.1f} us")

    # graphed end-to-end
    from stoix_amd.ops.graph import try_enable_graphs

    ok = try_enable_graphs(learner)
    print("graphs:", ok)
    t_step = timeit(lambda: learner.update_step(), iters=10)
    sps = (L.T * L.B) / (t_step / 1e3)
    print(f"graphed update_step: {t_step:8.2f} ms  ({sps/1e6:.2f}M steps/s)")


if __name__ == "__main__":
    main()
