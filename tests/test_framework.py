"""Config composition, checkpointing, logger, networks, running stats."""
import os

import numpy as np
import pytest
import torch

from stoix_amd.config import compose
from stoix_amd.networks.distributions import (
    AffineTanhTransformedDistribution,
    Categorical,
    EpsilonGreedy,
)
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace
from stoix_amd.ops import running_statistics as rs
from stoix_amd.utils.checkpointing import Checkpointer
from stoix_amd.utils.total_timestep_checker import check_total_timesteps


def test_compose_defaults_and_overrides():
    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/pendulum", "system.gamma=0.5", "arch.total_num_envs=64"],
    )
    assert cfg.env.scenario.name == "Pendulum-v1"
    assert cfg.system.gamma == 0.5
    assert cfg.arch.total_num_envs == 64
    assert cfg.logger.loggers == ["console", "json"]


def test_timestep_checker_derives_num_updates():
    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["arch.total_num_envs=32", "arch.total_timesteps=40960", "system.rollout_length=16"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    assert cfg.arch.num_envs == 32
    assert cfg.arch.num_updates == 40960 // (16 * 32)


def test_timestep_checker_rejects_bad_divisibility():
    cfg = compose("default/anakin/default_ff_ppo.yaml", ["arch.total_num_envs=7"])
    cfg.arch.n_devices = 2
    with pytest.raises(AssertionError):
        check_total_timesteps(cfg)


def test_checkpointer_roundtrip(tmp_path):
    ckpt = Checkpointer("test_model", {"cfg": 1}, directory=str(tmp_path))
    state = {"actor": {"w": torch.randn(3, 3)}, "step": torch.tensor(5)}
    ckpt.save(100, state, metric_value=1.0)
    template = {"actor": {"w": torch.zeros(3, 3)}, "step": torch.tensor(0)}
    restored = ckpt.restore_params(template)
    torch.testing.assert_close(restored["actor"]["w"], state["actor"]["w"])
    assert restored["step"].item() == 5


def test_checkpointer_best_retention(tmp_path):
    ckpt = Checkpointer("m", directory=str(tmp_path), max_to_keep=1)
    ckpt.save(1, {"w": torch.tensor([1.0])}, metric_value=5.0)
    ckpt.save(2, {"w": torch.tensor([2.0])}, metric_value=3.0)  # worse
    best = ckpt.restore_params({"w": torch.tensor([0.0])}, best=True)
    assert best["w"].item() == 1.0
    latest = ckpt.restore_params({"w": torch.tensor([0.0])})
    assert latest["w"].item() == 2.0


def test_checkpointer_version_gate(tmp_path):
    import json

    ckpt = Checkpointer("m", directory=str(tmp_path))
    ckpt.save(1, {"w": torch.tensor([1.0])})
    meta_path = os.path.join(str(tmp_path), "m", "metadata.json")
    with open(meta_path) as f:
        meta = json.load(f)
    meta["checkpointer_version"] = "1.0"
    with open(meta_path, "w") as f:
        json.dump(meta, f)
    with pytest.raises(ValueError):
        ckpt.restore_params({"w": torch.tensor([0.0])})


def test_running_statistics_welford_matches_numpy():
    g = torch.Generator().manual_seed(0)
    state = rs.init_state((5,))
    chunks = [torch.randn(100, 5, generator=g) * 3 + 1 for _ in range(4)]
    for c in chunks:
        state = rs.update(state, c)
    all_data = torch.cat(chunks)
    torch.testing.assert_close(state.mean, all_data.mean(0), rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(state.std, all_data.std(0, unbiased=False), rtol=1e-3, atol=1e-4)
    normed = rs.normalize(all_data, state)
    assert abs(normed.mean().item()) < 1e-4


def test_categorical_distribution():
    logits = torch.tensor([[2.0, 0.0, -1.0]])
    d = Categorical(logits)
    assert d.mode().item() == 0
    lp = d.log_prob(torch.tensor([0]))
    probs = torch.softmax(logits, -1)
    assert abs(lp.exp().item() - probs[0, 0].item()) < 1e-6
    ent = d.entropy()
    assert abs(ent.item() - (-(probs * probs.log()).sum()).item()) < 1e-5
    g = torch.Generator().manual_seed(0)
    samples = torch.stack([d.sample(g) for _ in range(2000)]).float()
    assert abs((samples == 0).float().mean().item() - probs[0, 0].item()) < 0.05


def test_tanh_normal_log_prob_consistency():
    d = AffineTanhTransformedDistribution(
        torch.zeros(4, 2), torch.ones(4, 2) * 0.5, -2.0, 2.0
    )
    a, lp = d.sample_and_log_prob(torch.Generator().manual_seed(1))
    assert (a.abs() <= 2.0).all()
    lp2 = d.log_prob(a)
    torch.testing.assert_close(lp, lp2, rtol=1e-3, atol=1e-3)
    assert d.mode().shape == (4, 2)


def test_tanh_normal_matches_torch_reference():
    """Cross-check log_prob against torch.distributions TransformedDistribution."""
    loc = torch.tensor([[0.3, -0.7]])
    scale = torch.tensor([[0.6, 1.2]])
    mine = AffineTanhTransformedDistribution(loc, scale, -1.0, 1.0)
    base = torch.distributions.Normal(loc, scale)
    ref = torch.distributions.TransformedDistribution(
        base, [torch.distributions.transforms.TanhTransform(cache_size=1)]
    )
    x = torch.tensor([[0.5, -0.3]])
    torch.testing.assert_close(
        mine.log_prob(x), ref.log_prob(x).sum(-1), rtol=1e-4, atol=1e-4
    )


def test_epsilon_greedy():
    prefs = torch.tensor([[1.0, 5.0, 2.0]])
    d = EpsilonGreedy(prefs, epsilon=0.3)
    assert d.mode().item() == 1
    g = torch.Generator().manual_seed(0)
    samples = torch.stack([d.sample(g) for _ in range(3000)])
    frac_greedy = (samples == 1).float().mean().item()
    assert abs(frac_greedy - (0.7 + 0.1)) < 0.05


def test_network_factory_builds_and_runs():
    obs = BoxSpace((8,))
    act_d = DiscreteSpace(4)
    act_c = BoxSpace((3,), -1.0, 1.0)
    net_cfg = {
        "pre_torso": {"_target_": "MLPTorso", "layer_sizes": [32, 32]},
        "action_head": {"_target_": "CategoricalHead"},
    }
    actor = build_actor(net_cfg, obs, act_d)
    dist = actor(torch.randn(5, 8))
    assert dist.sample().shape == (5,)
    net_cfg["action_head"] = {"_target_": "NormalAffineTanhDistributionHead"}
    actor_c = build_actor(net_cfg, obs, act_c)
    dist = actor_c(torch.randn(5, 8))
    assert dist.sample().shape == (5, 3)
    critic = build_critic(
        {"pre_torso": {"_target_": "MLPTorso", "layer_sizes": [32]}, "critic_head": {"_target_": "ScalarCriticHead"}},
        obs,
    )
    assert critic(torch.randn(5, 8)).shape == (5,)


def test_ppo_obs_normalization_reference_semantics():
    """Running statistics must be estimated from RAW observations (the
    trajectory is normalised post-rollout with pre-update stats; reference
    ff_ppo.py:148-162). A feedback loop on normalised obs would drive the
    stats to (0,1) regardless of the data."""
    import torch

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.ops import running_statistics as rs
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        ["env=brax/ant", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=4", "arch.num_evaluation=1",
         "system.rollout_length=8", "system.num_minibatches=2",
         "system.epochs=1", "system.normalize_observations=true",
         "system.fused=false", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(0)
    env = environments.make_single(cfg, 8, "cpu", seed=0)
    learner = PPOLearner(cfg, env, torch.device("cpu"))
    for _ in range(4):
        learner.update_step()
    # Ant's first obs dim is the torso z-height (~0.5, clearly non-zero):
    # raw-data statistics must capture it
    mean0 = float(learner.obs_stats.mean[0])
    assert 0.2 < mean0 < 0.9, f"stats look collapsed/raw-less: mean[0]={mean0}"
    # the stored trajectory is the NORMALISED one after rollout_phase
    learner.rollout_phase()
    assert float(learner.buf_obs.mean().abs()) < 1.0


def test_learner_checkpoint_roundtrip(tmp_path):
    """Full save -> fresh learner -> restore cycle: the restored learner's
    greedy actions match the original exactly (BASELINE checkpoint-format
    parity: the recovery story is restart-from-checkpoint)."""
    import torch

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.checkpointing import Checkpointer
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=2", "arch.num_evaluation=1", "system.rollout_length=8",
         "system.num_minibatches=2", "system.epochs=1", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(0)
    env = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner = PPOLearner(cfg, env, torch.device("cpu"))
    for _ in range(2):
        learner.update_step()
    ckpt = Checkpointer("rt", {"algo": "ppo"}, directory=str(tmp_path))
    ckpt.save(16, learner.state_for_checkpoint(), metric_value=1.0)

    torch.manual_seed(123)  # different init
    env2 = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner2 = PPOLearner(cfg, env2, torch.device("cpu"))
    restored = ckpt.restore_params(learner2.state_for_checkpoint())
    learner2.load_params(restored)

    obs = torch.randn(16, 4)
    a1 = learner.act_fn(obs, True)
    a2 = learner2.act_fn(obs, True)
    torch.testing.assert_close(a1, a2)


def test_scanned_rnn_hoisted_matches_per_step():
    """The hoisted-GEMM training scan (input projection for all T in one
    GEMM) must match the generic per-step cell path bit-closely for GRU
    and LSTM, including done-masked resets (K13 torch side)."""
    import torch

    from stoix_amd.networks.base import ScannedRNN

    torch.manual_seed(0)
    for kind in ("gru", "lstm"):
        rnn = ScannedRNN(12, 32, cell_type=kind)
        T, B = 9, 7
        x = torch.randn(T, B, 12)
        resets = torch.rand(T, B) < 0.3
        st0 = rnn.initial_state(B, "cpu")
        out_fast, st_fast = rnn(x, resets, list(st0))
        orig = rnn._single_cell
        rnn._single_cell = lambda: (None, None)  # force generic path
        out_ref, st_ref = rnn(x, resets, list(st0))
        rnn._single_cell = orig
        torch.testing.assert_close(out_fast, out_ref, rtol=1e-5, atol=1e-6)
        if kind == "gru":
            torch.testing.assert_close(st_fast[0], st_ref[0], rtol=1e-5, atol=1e-6)
        else:
            torch.testing.assert_close(st_fast[0][0], st_ref[0][0], rtol=1e-5, atol=1e-6)
            torch.testing.assert_close(st_fast[0][1], st_ref[0][1], rtol=1e-5, atol=1e-6)


def test_running_statistics_update_inplace_matches_functional():
    """rs.update_ must leave the state's ORIGINAL tensors holding the same
    values as the functional update (the fused engine's kernels hold those
    addresses across hip-graph replays)."""
    import torch

    from stoix_amd.ops import running_statistics as rs

    torch.manual_seed(0)
    s1 = rs.init_state((5,))
    s2 = rs.init_state((5,))
    mean_ptr = s2.mean.data_ptr()
    for _ in range(3):
        batch = torch.randn(64, 5) * 3 + 1
        s1 = rs.update(s1, batch)
        rs.update_(s2, batch)
    assert s2.mean.data_ptr() == mean_ptr  # stable address
    torch.testing.assert_close(s1.mean, s2.mean)
    torch.testing.assert_close(s1.std, s2.std)
    torch.testing.assert_close(s1.count, s2.count)


def test_all_network_presets_build():
    """Every shipped network preset yaml builds an actor (and runs a
    forward) against an appropriate space — mirrors the reference's config
    surface (configs/network/*)."""
    import os

    import torch

    from stoix_amd.config import CONFIG_ROOT, compose
    from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace
    from stoix_amd.networks.factory import build_actor

    pixel = {"cnn", "visual_resnet"}
    skip = {"world_model", "mlp_rainbow", "mlp_dueling_dqn"}  # built by their systems directly
    rnn = {"rnn", "rnn_dqn", "rnn_dueling_dqn"}
    cont = {"mlp_continuous", "mlp_sac", "mlp_ddpg", "mlp_d4pg", "mlp_mpo_continuous"}
    for f in sorted(os.listdir(CONFIG_ROOT / "network")):
        name = f[:-5]
        if name in skip:
            continue
        cfg = compose("default/anakin/default_ff_ppo.yaml", [f"network={name}"])
        obs = BoxSpace((84, 84, 1) if name in pixel else (8,), 0.0, 1.0)
        if name in cont:
            act = BoxSpace((3,), -1.0, 1.0)
        else:
            act = DiscreteSpace(4)
        if name in rnn:
            from stoix_amd.networks.factory import build_recurrent_actor

            a = build_recurrent_actor(cfg.network.actor_network, obs, act)
            st = a.initial_state(2, "cpu")
            x = torch.zeros(3, 2, *obs.shape)
            resets = torch.zeros(3, 2, dtype=torch.bool)
            out, _ = a(x, resets, st)
        else:
            a = build_actor(cfg.network.actor_network, obs, act)
            out = a(torch.zeros(2, *obs.shape))
        assert out is not None, name


def test_tanh_normal_boundary_log_prob_bounded():
    """Reference boundary semantics (distributions.py:55-80): saturated
    distributions report the log of the squashed MASS (bounded, ~+6.9 per
    dim at eps=1e-3), NOT the clamped-atanh density (which reached -900
    and blew up PPO ratios — found by the swing-up stress test)."""
    import torch

    from stoix_amd.networks.distributions import AffineTanhTransformedDistribution

    loc = torch.tensor([[-40.0, 40.0, 0.0]])
    scale = torch.tensor([[1e-3, 1e-3, 0.5]])
    d = AffineTanhTransformedDistribution(loc, scale, -1.0, 1.0)
    a = d.sample()
    lp = d.log_prob(a)
    assert torch.isfinite(lp).all()
    assert float(lp) > -20.0  # bounded, not -900
    # interior values still use the exact change-of-variables density
    d2 = AffineTanhTransformedDistribution(torch.zeros(1, 3), torch.full((1, 3), 0.6), -1.0, 1.0)
    a2, lp_s = d2.sample_and_log_prob()
    torch.testing.assert_close(d2.log_prob(a2), lp_s, rtol=1e-4, atol=1e-4)
    # gradients flow through the boundary branches (log_ndtr is smooth)
    loc3 = torch.nn.Parameter(torch.tensor([[5.0]]))
    d3 = AffineTanhTransformedDistribution(loc3, torch.full((1, 1), 0.1), -1.0, 1.0)
    d3.log_prob(torch.tensor([[1.0]])).backward()
    assert torch.isfinite(loc3.grad).all() and float(loc3.grad.abs()) > 0


def test_tanh_normal_density_mass_accounting():
    """Exact mass accounting for the tanh-normal with the CDF-mass
    boundary branches: (a) interior densities equal the analytic
    change-of-variables density, (b) the boundary values equal
    tail-mass / eps, (c) tails + interior mass sum to 1 — including the
    saturated regime where nearly ALL mass sits in a boundary strip."""
    import math

    import torch

    from stoix_amd.networks.distributions import AffineTanhTransformedDistribution

    eps = 1e-3
    u_hi = math.atanh(1.0 - eps)
    for loc_v, scale_v in [(0.0, 0.6), (1.5, 0.3), (6.0, 0.5), (-40.0, 1e-3)]:
        d = AffineTanhTransformedDistribution(
            torch.tensor([[loc_v]]), torch.tensor([[scale_v]]), -1.0, 1.0
        )
        # (a) pointwise interior density == N(u; loc, scale) / (1 - a^2)
        xs = torch.linspace(-0.99, 0.99, 41).view(-1, 1)
        u = torch.atanh(xs)
        z = (u - loc_v) / scale_v
        ref = (
            torch.exp(-0.5 * z * z) / (scale_v * math.sqrt(2 * math.pi))
            / (1 - xs * xs)
        ).view(-1)
        got = d.log_prob(xs).exp()
        torch.testing.assert_close(got, ref, rtol=1e-3, atol=1e-8)
        # (b) boundary value * eps == analytic tail mass
        zr = torch.tensor((u_hi - loc_v) / scale_v, dtype=torch.float64)
        zl = torch.tensor((-u_hi - loc_v) / scale_v, dtype=torch.float64)
        right_mass = float(torch.special.ndtr(-zr))
        left_mass = float(torch.special.ndtr(zl))
        got_right = float(d.log_prob(torch.tensor([[1.0]])).exp() * eps)
        got_left = float(d.log_prob(torch.tensor([[-1.0]])).exp() * eps)
        assert abs(got_right - right_mass) < 1e-4 + 0.01 * right_mass
        assert abs(got_left - left_mass) < 1e-4 + 0.01 * left_mass
        # (c) tails + interior mass == 1
        interior_mass = float(torch.special.ndtr(zr) - torch.special.ndtr(zl))
        assert abs(interior_mass + left_mass + right_mass - 1.0) < 1e-9


def test_distribution_numerics_vs_scipy_and_torch():
    """Golden numerics for the distribution layer against independent
    references: scipy.stats for Categorical entropy/KL and Beta log-pdf,
    torch.distributions closed forms for Normal/MVN-diag entropy, log_prob
    and KL (the quantities MPO's dual constraints and PPO's entropy bonus
    are computed from)."""
    import scipy.stats as ss

    from stoix_amd.networks import distributions as D

    g = torch.Generator().manual_seed(0)
    logits = torch.randn(5, 7, generator=g)
    logits2 = torch.randn(5, 7, generator=g)
    cat, cat2 = D.Categorical(logits), D.Categorical(logits2)
    p = torch.softmax(logits, -1).numpy()
    q = torch.softmax(logits2, -1).numpy()
    for i in range(5):
        assert abs(float(cat.entropy()[i]) - ss.entropy(p[i])) < 1e-5
        assert (
            abs(float(cat.kl_divergence(cat2)[i]) - ss.entropy(p[i], q[i])) < 1e-5
        )

    loc = torch.randn(6, 3, generator=g)
    scale = torch.rand(6, 3, generator=g) + 0.1
    x = torch.randn(6, 3, generator=g)
    tn = torch.distributions.Normal(loc, scale)
    n = D.Normal(loc, scale)
    torch.testing.assert_close(n.log_prob(x), tn.log_prob(x), rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(n.entropy(), tn.entropy(), rtol=1e-6, atol=1e-6)

    mvn = D.MultivariateNormalDiag(loc, scale)
    tmvn = torch.distributions.Independent(torch.distributions.Normal(loc, scale), 1)
    torch.testing.assert_close(mvn.log_prob(x), tmvn.log_prob(x), rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(mvn.entropy(), tmvn.entropy(), rtol=1e-6, atol=1e-6)

    alpha = torch.rand(8, 2, generator=g) * 3 + 0.5
    beta = torch.rand(8, 2, generator=g) * 3 + 0.5
    b = D.ClippedBeta(alpha, beta)
    v = torch.rand(8, 2, generator=g).clamp(1e-3, 1 - 1e-3)
    # event dim is the last axis: log_prob sums the per-component Beta pdfs
    ref = torch.tensor(
        ss.beta.logpdf(v.numpy(), alpha.numpy(), beta.numpy()), dtype=torch.float32
    ).sum(-1)
    torch.testing.assert_close(b.log_prob(v), ref, rtol=1e-4, atol=1e-5)


def test_checkpoint_resume_restores_optimizer_moments(tmp_path):
    """True-resume parity: the reference checkpoints the whole learner
    state including opt_states. The aux payload must round-trip the Adam
    moments so a restored learner continues with the same optimizer
    trajectory (not freshly-zeroed moments)."""
    import torch

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd.utils.checkpointing import Checkpointer
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=2", "arch.num_evaluation=1", "system.rollout_length=8",
         "system.num_minibatches=2", "system.epochs=1", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(0)
    env = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner = PPOLearner(cfg, env, torch.device("cpu"))
    for _ in range(2):
        learner.update_step()
    ckpt = Checkpointer("resume", {}, directory=str(tmp_path))
    ckpt.stage_aux(learner.aux_checkpoint_state())
    ckpt.save(16, learner.state_for_checkpoint(), metric_value=1.0)

    torch.manual_seed(99)
    env2 = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner2 = PPOLearner(cfg, env2, torch.device("cpu"))
    learner2.load_params(ckpt.restore_params(learner2.state_for_checkpoint()))
    aux = ckpt.restore_aux()
    assert aux is not None
    learner2.load_aux_checkpoint_state(aux)
    s1 = learner.actor_opt.state_dict()["state"]
    s2 = learner2.actor_opt.state_dict()["state"]
    assert set(s1) == set(s2) and len(s1) > 0
    for k in s1:
        torch.testing.assert_close(s1[k]["exp_avg"], s2[k]["exp_avg"])
        torch.testing.assert_close(s1[k]["exp_avg_sq"], s2[k]["exp_avg_sq"])


def test_default_aux_state_covers_all_learner_optimizers():
    """The generic aux payload discovers every torch Optimizer on a
    learner (SAC holds three) and restores moments by attribute name."""
    import torch

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.anakin import default_aux_state, load_default_aux_state
    from stoix_amd.systems.sac.ff_sac import SACLearner
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_sac.yaml",
        ["env=classic/pendulum", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=2", "arch.num_evaluation=1", "system.rollout_length=4",
         "system.batch_size=16", "system.buffer_size=512", "system.warmup_steps=16",
         "system.epochs=1", "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(0)
    env = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner = SACLearner(cfg, env, torch.device("cpu"))
    for _ in range(2):
        learner.update_step()
    aux = default_aux_state(learner)
    assert {"actor_opt", "q_opt", "alpha_opt"} <= set(aux)
    torch.manual_seed(7)
    env2 = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner2 = SACLearner(cfg, env2, torch.device("cpu"))
    for _ in range(2):
        learner2.update_step()  # populate slots with DIFFERENT moments
    load_default_aux_state(learner2, aux)
    s1 = learner.q_opt.state_dict()["state"]
    s2 = learner2.q_opt.state_dict()["state"]
    for k in s1:
        torch.testing.assert_close(s1[k]["exp_avg"], s2[k]["exp_avg"])


def test_noisy_linear_statistics_and_semantics():
    """NoisyLinear (Fortunato et al. factorised noise): with mu=0, the
    output's std over resamples matches sigma * E|f(eps_out)f(eps_in)| ...
    checked empirically: Var[y_j] = sum_i sigma^2 E[(f_out f_in)^2] x_i^2 +
    sigma_b^2 E[f_out^2], with E[f(eps)^2] = E[|eps|] = sqrt(2/pi).
    Also: noise-off equals the plain linear; materialize caches the draw."""
    import math

    import torch

    from stoix_amd.networks.layers import NoisyLinear

    torch.manual_seed(0)
    lin = NoisyLinear(32, 16, sigma_zero=0.5)
    with torch.no_grad():
        lin.weight_mu.zero_()
        lin.bias_mu.zero_()
    x = torch.randn(1, 32)
    g = torch.Generator().manual_seed(1)
    outs = []
    for _ in range(4000):
        lin.resample_noise(g)
        outs.append(lin(x).detach())
    y = torch.cat(outs)  # [4000, 16]
    assert abs(float(y.mean())) < 0.02  # E[f(eps)] = 0
    sig = 0.5 / math.sqrt(32)
    e_abs = math.sqrt(2.0 / math.pi)  # E[f(eps)^2] = E|eps|
    expect_var = (sig**2) * (e_abs**2) * float((x**2).sum()) + (sig**2) * e_abs
    assert abs(float(y.var()) / expect_var - 1.0) < 0.15, (float(y.var()), expect_var)

    # noise off == plain linear on the mus
    with torch.no_grad():
        lin.weight_mu.normal_()
        lin.bias_mu.normal_()
    lin.use_noise = False
    torch.testing.assert_close(lin(x), torch.nn.functional.linear(x, lin.weight_mu, lin.bias_mu))
    lin.use_noise = True

    # materialize caches the current draw: same output until resample
    lin.resample_noise(g)
    lin.materialize()
    a = lin(x).detach()
    b = lin(x).detach()
    torch.testing.assert_close(a, b)
    lin._mat = None
    lin.resample_noise(g)
    c = lin(x).detach()
    assert not torch.allclose(a, c)


def test_mpo_temperature_dual_enforces_kl_budget():
    """MPO E-step property (Abdolmaleki et al. 2018): minimising the
    temperature dual  g(eta) = eta*eps + eta*mean(logsumexp(Q/eta) - log M)
    yields weights softmax(Q/eta*) whose average KL to the uniform
    proposal equals the budget eps. Verified numerically on random Q with
    the exact expressions ff_mpo's temp_loss uses, via autograd descent
    on eta (the dual is convex in eta)."""
    import math

    import torch

    g = torch.Generator().manual_seed(0)
    M, B = 32, 16
    q = torch.randn(M, B, generator=g) * 2.0
    eps = 0.1
    log_eta = torch.tensor(0.0, requires_grad=True)
    opt = torch.optim.Adam([log_eta], lr=0.05)
    for _ in range(800):
        eta = torch.nn.functional.softplus(log_eta) + 1e-8
        dual = eta * (eps + (torch.logsumexp(q / eta, dim=0) - math.log(M)).mean())
        opt.zero_grad()
        dual.backward()
        opt.step()
    eta = float(torch.nn.functional.softplus(log_eta.detach()))
    w = torch.softmax(q / eta, dim=0)
    # KL(w || uniform) per batch column, averaged
    kl = (w * (w.clamp_min(1e-12).log() + math.log(M))).sum(0).mean()
    assert abs(float(kl) - eps) < 0.02, (float(kl), eps)


def test_mpo_decoupled_mstep_gradient_structure():
    """MPO's decoupled M-step: the mean CE term must carry NO gradient to
    sigma and the std CE term NONE to mu (each uses the TARGET value of
    the other parameter) — the structural property that lets the KL duals
    constrain mean and std independently."""
    import math

    import torch

    g = torch.Generator().manual_seed(1)
    M, B, D = 8, 4, 3
    mu_t = torch.randn(B, D, generator=g)
    sigma_t = torch.rand(B, D, generator=g) + 0.5
    a = mu_t.unsqueeze(0) + sigma_t.unsqueeze(0) * torch.randn(M, B, D, generator=g)
    mu_o = torch.randn(B, D, generator=g).requires_grad_(True)
    sigma_o = (torch.rand(B, D, generator=g) + 0.5).requires_grad_(True)

    def normal_logp(mu, sigma, x):
        var = sigma**2
        return (-((x - mu) ** 2) / (2 * var) - sigma.log() - 0.5 * math.log(2 * math.pi)).sum(-1)

    lp_mean = normal_logp(mu_o.unsqueeze(0), sigma_t.unsqueeze(0), a).sum()
    gm, gs = torch.autograd.grad(lp_mean, [mu_o, sigma_o], allow_unused=True)
    assert gs is None and gm is not None and gm.abs().sum() > 0
    lp_std = normal_logp(mu_t.unsqueeze(0), sigma_o.unsqueeze(0), a).sum()
    gm, gs = torch.autograd.grad(lp_std, [mu_o, sigma_o], allow_unused=True)
    assert gm is None and gs is not None and gs.abs().sum() > 0


def test_spo_systematic_resample_properties():
    """Systematic resampling (SPO's SMC backbone): (a) unbiased — the
    resampled count of particle i is within 1 of P*w_i for EVERY draw
    (the defining low-variance property of systematic resampling),
    (b) expectation-preserving — weighted mean of values before equals
    the unweighted mean after, up to that +-1/P quantisation."""
    import torch

    from stoix_amd.systems.spo.ff_spo import systematic_resample

    g = torch.Generator().manual_seed(0)
    B, P = 6, 64
    logw = torch.randn(B, P, generator=g) * 2.0
    w = torch.softmax(logw, dim=-1)
    vals = torch.randn(B, P, generator=g)
    for trial in range(20):
        idx = systematic_resample(logw, g)
        counts = torch.zeros(B, P)
        counts.scatter_add_(1, idx, torch.ones(B, P))
        # (a) deterministic-within-1 counts
        assert torch.all((counts - P * w).abs() <= 1.0 + 1e-5), trial
        # (b) post-resample mean ~= weighted mean: counts/P is within 1/P
        # of w_i per particle, so |post - pre| <= (1/P) * sum_i |v_i|
        post = (counts / P * vals).sum(-1)
        pre = (w * vals).sum(-1)
        bound = vals.abs().sum(-1) / P + 1e-6
        assert torch.all((post - pre).abs() <= bound), trial


def test_scale_gradient_halves_backward_not_forward():
    """MuZero's scale_gradient(x, 0.5) (reference jax_utils.py:12 /
    ff_mz unroll): forward is identity, backward scales by 0.5 — and a
    k-step unroll through it scales the state gradient by 0.5^k."""
    import torch

    from stoix_amd.systems.search.ff_mz import scale_gradient

    x = torch.randn(5, requires_grad=True)
    y = scale_gradient(x, 0.5)
    torch.testing.assert_close(y, x)
    (g,) = torch.autograd.grad(y.sum(), x)
    torch.testing.assert_close(g, torch.full_like(x, 0.5))
    # chained: 3 unroll steps -> 0.5^3
    z = x
    for _ in range(3):
        z = scale_gradient(z * 1.0, 0.5)
    (g,) = torch.autograd.grad(z.sum(), x)
    torch.testing.assert_close(g, torch.full_like(x, 0.125))


def test_r2d2_value_transform_pair_is_inverse():
    """R2D2's signed hyperbolic h and signed parabolic h^-1 (Pohlen et al.
    transform pair) must be exact inverses over a wide range, odd, and
    monotone — the properties the TD targets rely on."""
    import torch

    from stoix_amd.systems.q_learning.rec_r2d2 import (
        signed_hyperbolic,
        signed_parabolic,
    )

    x = torch.linspace(-500.0, 500.0, 2001, dtype=torch.float64)
    h = signed_hyperbolic(x)
    back = signed_parabolic(h)
    torch.testing.assert_close(back, x, rtol=1e-6, atol=1e-6)
    assert abs(float(signed_hyperbolic(torch.zeros(1, dtype=torch.float64)))) < 1e-12
    torch.testing.assert_close(signed_hyperbolic(-x), -h)
    assert torch.all(h[1:] > h[:-1])  # strictly increasing
    # compresses: |h(x)| grows ~sqrt, so h(400) << 400
    assert float(h[-1]) < 25.0


def test_two_hot_encoding_properties():
    """two_hot (MuZero/DisCo scalar targets): mass 1, expectation equals
    the (clamped) input scalar, at most two adjacent non-zeros, exact
    one-hot on atom values."""
    import torch

    from stoix_amd.networks.model_based import two_hot

    atoms = torch.linspace(-5.0, 5.0, 21)
    g = torch.Generator().manual_seed(0)
    x = torch.rand(64, generator=g) * 14.0 - 7.0  # includes out-of-range
    enc = two_hot(x, atoms)
    torch.testing.assert_close(enc.sum(-1), torch.ones(64))
    torch.testing.assert_close((enc * atoms).sum(-1), x.clamp(-5.0, 5.0), rtol=1e-5, atol=1e-6)
    nz = (enc > 1e-8).sum(-1)
    assert torch.all(nz <= 2)
    # adjacency: the two mass-bearing bins are neighbours
    for b in range(64):
        on = (enc[b] > 1e-8).nonzero().flatten()
        if on.numel() == 2:
            assert int(on[1] - on[0]) == 1, (b, on)
    # exact atoms -> one-hot
    enc2 = two_hot(atoms.clone(), atoms)
    torch.testing.assert_close(enc2, torch.eye(21), rtol=0, atol=1e-6)


def test_discrete_valued_and_multidiscrete_distributions():
    """DiscreteValuedDistribution (D4PG critic): mean = sum p*support and
    mode = support[argmax]. MultiDiscreteDistribution: log_prob factorises
    as the sum of per-dimension categorical log-probs."""
    import torch

    from stoix_amd.networks.distributions import (
        Categorical,
        DiscreteValuedDistribution,
        MultiDiscreteDistribution,
    )

    g = torch.Generator().manual_seed(0)
    logits = torch.randn(4, 11, generator=g)
    support = torch.linspace(-2.0, 2.0, 11)
    d = DiscreteValuedDistribution(logits, support)
    p = torch.softmax(logits, -1)
    torch.testing.assert_close(d.mean(), (p * support).sum(-1))
    torch.testing.assert_close(d.mode(), support[logits.argmax(-1)])

    nv = [3, 4, 2]
    flat = torch.randn(5, sum(nv), generator=g)
    md = MultiDiscreteDistribution(flat, nv)
    a = torch.stack(
        [torch.randint(0, n, (5,), generator=g) for n in nv], dim=-1
    )
    lp = md.log_prob(a)
    expect = torch.zeros(5)
    off = 0
    for i, n in enumerate(nv):
        expect = expect + Categorical(flat[:, off : off + n]).log_prob(a[:, i])
        off += n
    torch.testing.assert_close(lp, expect)


def test_dueling_decomposition_identity():
    """Dueling Q nets: mean-centred advantages imply mean_a Q(s,a) = V(s)
    — the identifiability constraint that motivates the architecture —
    and the distributional variant satisfies it per atom (before the
    softmax)."""
    import torch

    from stoix_amd.networks.dueling import (
        DistributionalDuelingQNetwork,
        DuelingQNetwork,
    )

    torch.manual_seed(0)
    net = DuelingQNetwork(6, 4, layer_sizes=(32,))
    x = torch.randn(10, 6)
    q = net.q_values(x)
    v = net.value(x).squeeze(-1)
    torch.testing.assert_close(q.mean(-1), v, rtol=1e-5, atol=1e-5)

    dnet = DistributionalDuelingQNetwork(6, 4, num_atoms=11, layer_sizes=(32,))
    out = dnet(x)
    logits = out.logits if hasattr(out, "logits") else out.q_logits
    assert logits.shape == (10, 4, 11)
    # per-atom identifiability: mean over actions of the dueled atom
    # logits equals the value stream's atom logits
    v_atoms = dnet.value(x)
    torch.testing.assert_close(logits.mean(dim=1), v_atoms, rtol=1e-5, atol=1e-5)


def test_run_restores_from_checkpoint_at_startup(tmp_path):
    """logger.checkpointing.load_model parity (reference ff_ppo.py:504-512):
    a run pointed at a saved checkpoint starts from ITS params (and aux
    optimizer moments), not from fresh init — and loading must not clobber
    the saved metadata/version gate."""
    import json

    import torch

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner, run
    from stoix_amd.utils.checkpointing import Checkpointer
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=2", "arch.num_evaluation=1", "arch.num_eval_episodes=2",
         "system.rollout_length=8", "system.num_minibatches=2", "system.epochs=1",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    torch.manual_seed(0)
    env = environments.make_single(cfg, 8, torch.device("cpu"), seed=0)
    learner = PPOLearner(cfg, env, torch.device("cpu"))
    for _ in range(2):
        learner.update_step()
    ckpt = Checkpointer("ff_ppo", {"src": "test"}, directory=str(tmp_path))
    ckpt.stage_aux(learner.aux_checkpoint_state())
    ckpt.save(16, learner.state_for_checkpoint(), metric_value=1.0)
    saved_w = {k: v.clone() for k, v in learner.actor.state_dict().items()}

    cfg2 = compose(
        "default/anakin/default_ff_ppo.yaml",
        ["env=classic/cartpole", "arch.total_num_envs=8", "arch.total_timesteps=null",
         "arch.num_updates=1", "arch.num_evaluation=1", "arch.num_eval_episodes=2",
         "arch.absolute_metric=false",
         "system.rollout_length=8", "system.num_minibatches=2", "system.epochs=1",
         "logger.loggers=[]", "logger.checkpointing.save_model=false",
         "logger.checkpointing.load_model=true",
         f"logger.checkpointing.load_args.checkpoint_uid={tmp_path}"],
    )
    # run() trains 1 more update from the restored weights; the restore
    # itself is verified by rebuilding the learner the same way run does
    # and catching the loaded weights before any update
    import stoix_amd.systems.anakin as anakin_mod

    captured = {}
    orig = PPOLearner.update_step

    def spy(self):
        if "w" not in captured:
            captured["w"] = {k: v.clone() for k, v in self.actor.state_dict().items()}
        return orig(self)

    PPOLearner.update_step = spy
    try:
        run(cfg2)
    finally:
        PPOLearner.update_step = orig
    for k in saved_w:
        torch.testing.assert_close(captured["w"][k], saved_w[k])
    # metadata survived the loader
    meta = json.loads((tmp_path / "ff_ppo" / "metadata.json").read_text())
    assert meta.get("src") == "test"


def test_checkpointer_keep_period_and_save_interval(tmp_path):
    """save_args parity: keep_period protects periodic checkpoints from
    max_to_keep GC, and save_interval_steps thins the per-eval saves."""
    ckpt = Checkpointer("m", {}, directory=str(tmp_path), max_to_keep=1, keep_period=20)
    for t in (10, 20, 30, 40):
        ckpt.save(t, {"w": torch.tensor([float(t)])})
    kept = sorted(
        int(d.split("_")[1]) for d in os.listdir(tmp_path / "m") if d.startswith("step_")
    )
    # 20 and 40 survive via keep_period (multiples of 20); 40 is also latest
    assert kept == [20, 40], kept


def test_polyak_update_foreach_semantics():
    """polyak_update(src, dst, tau): dst <- tau*src + (1-tau)*dst, exactly,
    src untouched (the _foreach fused helper every off-policy system's
    target nets rely on)."""
    import torch

    from stoix_amd.parallel.dist import polyak_update

    torch.manual_seed(0)
    src = torch.nn.Linear(6, 5)
    dst = torch.nn.Linear(6, 5)
    src_before = [p.clone() for p in src.parameters()]
    dst_before = [p.clone() for p in dst.parameters()]
    tau = 0.25
    with torch.no_grad():
        polyak_update(src.parameters(), dst.parameters(), tau)
    for p, pb in zip(src.parameters(), src_before):
        torch.testing.assert_close(p, pb)  # source unchanged
    for p, sb, db in zip(dst.parameters(), src_before, dst_before):
        torch.testing.assert_close(p, tau * sb + (1 - tau) * db)
