"""GPU numerics tests: every HIP kernel vs its plain-PyTorch fp32 reference
(the CPU implementations in stoix_amd.ops / stoix_amd.envs)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@pytest.fixture(scope="module")
def ext():
    from stoix_amd import ops

    e = ops.ext(required=True)
    assert e is not None
    return e


@requires_gpu
def test_gae_kernel_matches_reference(ext):
    from stoix_amd.ops import multistep as ms

    g = torch.Generator().manual_seed(0)
    T, B = 37, 513
    r = torch.randn(T, B, generator=g)
    d = (torch.rand(T, B, generator=g) > 0.1).float() * 0.99
    v = torch.randn(T, B, generator=g)
    vb = torch.randn(T, B, generator=g)
    tr = torch.rand(T, B, generator=g) > 0.9
    adv_cpu, tgt_cpu = ms.batch_truncated_generalized_advantage_estimation(
        r, d, 0.95, v, vb, truncation_t=tr
    )
    adv_gpu, tgt_gpu = ms.batch_truncated_generalized_advantage_estimation(
        r.cuda(), d.cuda(), 0.95, v.cuda(), vb.cuda(), truncation_t=tr.cuda()
    )
    torch.testing.assert_close(adv_gpu.cpu(), adv_cpu, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(tgt_gpu.cpu(), tgt_cpu, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_lambda_returns_kernel(ext):
    from stoix_amd.ops import multistep as ms

    g = torch.Generator().manual_seed(1)
    T, B = 16, 300
    r = torch.randn(T, B, generator=g)
    d = torch.full((T, B), 0.97)
    v = torch.randn(T, B, generator=g)
    cpu = ms.batch_lambda_returns(r, d, v, 0.9)
    gpu = ms.batch_lambda_returns(r.cuda(), d.cuda(), v.cuda(), 0.9)
    torch.testing.assert_close(gpu.cpu(), cpu, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_vtrace_kernel(ext):
    from stoix_amd.ops import multistep as ms

    g = torch.Generator().manual_seed(2)
    T, B = 20, 128
    v_tm1 = torch.randn(T, B, generator=g)
    v_t = torch.randn(T, B, generator=g)
    r = torch.randn(T, B, generator=g)
    d = torch.full((T, B), 0.99)
    rho = (torch.randn(T, B, generator=g) * 0.3).exp()
    e_cpu, pg_cpu, q_cpu = ms.vtrace_td_error_and_advantage(v_tm1, v_t, r, d, rho)
    e_gpu, pg_gpu, q_gpu = ms.vtrace_td_error_and_advantage(
        v_tm1.cuda(), v_t.cuda(), r.cuda(), d.cuda(), rho.cuda()
    )
    torch.testing.assert_close(e_gpu.cpu(), e_cpu, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(pg_gpu.cpu(), pg_cpu, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_retrace_kernel(ext):
    from stoix_amd.ops import multistep as ms

    g = torch.Generator().manual_seed(3)
    T, B = 12, 64
    q = torch.randn(T, B, generator=g)
    v = torch.randn(T, B, generator=g)
    r = torch.randn(T, B, generator=g)
    d = torch.full((T, B), 0.95)
    lr_ = torch.randn(T, B, generator=g) * 0.5
    cpu = ms.batch_retrace_continuous(q, q, v, r, d, lr_, 0.9)
    gpu = ms.batch_retrace_continuous(
        q.cuda(), q.cuda(), v.cuda(), r.cuda(), d.cuda(), lr_.cuda(), 0.9
    )
    torch.testing.assert_close(gpu.cpu(), cpu, rtol=1e-4, atol=1e-4)


@requires_gpu
def test_fused_adam_matches_torch(ext):
    n = 100_003
    g = torch.Generator().manual_seed(4)
    p0 = torch.randn(n, generator=g)
    grad = torch.randn(n, generator=g)

    # torch reference: clip-by-global-norm + Adam, 3 steps
    p_ref = torch.nn.Parameter(p0.clone())
    opt = torch.optim.Adam([p_ref], lr=1e-3, betas=(0.9, 0.999), eps=1e-8)
    for _ in range(3):
        opt.zero_grad()
        p_ref.grad = grad.clone()
        torch.nn.utils.clip_grad_norm_([p_ref], 0.5)
        opt.step()

    # HIP fused path
    p = p0.clone().cuda()
    gr = grad.clone().cuda()
    m = torch.zeros(n).cuda()
    v = torch.zeros(n).cuda()
    sqnorm = torch.zeros(1).cuda()
    step_t = torch.zeros(1, dtype=torch.long).cuda()
    for _ in range(3):
        ext.fused_adam(p, gr, m, v, sqnorm, step_t, 1e-3, 0.9, 0.999, 1e-8, 0.5)
    torch.cuda.synchronize()
    torch.testing.assert_close(p.cpu(), p_ref.detach(), rtol=1e-5, atol=1e-6)


@requires_gpu
def test_polyak_kernel(ext):
    g = torch.Generator().manual_seed(5)
    online = torch.randn(1000, generator=g).cuda()
    target = torch.randn(1000, generator=g).cuda()
    expected = 0.01 * online + 0.99 * target
    ext.polyak(online, target, 0.01)
    torch.cuda.synchronize()
    torch.testing.assert_close(target, expected, rtol=1e-6, atol=1e-7)


@requires_gpu
def test_cartpole_kernel_matches_torch_env(ext):
    """Physics parity: same state + action on CPU-torch and GPU-HIP paths."""
    from stoix_amd.envs.classic import CartPole

    cpu_env = CartPole(64, device="cpu", seed=0)
    gpu_env = CartPole(64, device="cuda", seed=0)
    assert gpu_env._hip is not None
    ts_c = cpu_env.reset()
    # force identical state
    gpu_env.reset()
    gpu_env._state["s"].copy_(cpu_env._state["s"].cuda())
    for i in range(30):
        a = torch.randint(0, 2, (64,))
        ts_c = cpu_env.step(a)
        ts_g = gpu_env.step(a.cuda())
        # compare only envs that have not autoreset (reset noise differs)
        alive = ~ts_c.extras["episode_metrics"]["is_terminal_step"]
        alive_g = ~ts_g.extras["episode_metrics"]["is_terminal_step"].cpu()
        torch.testing.assert_close(alive, alive_g)
        torch.testing.assert_close(
            ts_g.observation.cpu()[alive], ts_c.observation[alive], rtol=1e-4, atol=1e-5
        )
        torch.testing.assert_close(ts_g.reward.cpu(), ts_c.reward)
        torch.testing.assert_close(ts_g.discount.cpu(), ts_c.discount)
        torch.testing.assert_close(ts_g.step_type.cpu(), ts_c.step_type)
        # true-final-obs parity on every env (written before autoreset)
        torch.testing.assert_close(
            ts_g.extras["next_obs"].cpu(), ts_c.extras["next_obs"], rtol=1e-4, atol=1e-5
        )
        if bool(ts_c.last().any()):
            # resync state after divergent reset noise
            gpu_env._state["s"].copy_(cpu_env._state["s"].cuda())


@requires_gpu
def test_ant_kernel_matches_torch_env(ext):
    from stoix_amd.envs.ant import Ant

    cpu_env = Ant(32, device="cpu", seed=0)
    gpu_env = Ant(32, device="cuda", seed=0)
    assert gpu_env._hip is not None
    cpu_env.reset()
    gpu_env.reset()
    gpu_env._state["s"].copy_(cpu_env._state["s"].cuda())
    g = torch.Generator().manual_seed(7)
    for i in range(20):
        a = torch.rand(32, 8, generator=g) * 2 - 1
        ts_c = cpu_env.step(a)
        ts_g = gpu_env.step(a.cuda())
        alive = ~ts_c.extras["episode_metrics"]["is_terminal_step"]
        torch.testing.assert_close(
            ts_g.extras["next_obs"].cpu(), ts_c.extras["next_obs"], rtol=2e-3, atol=2e-4
        )
        torch.testing.assert_close(ts_g.reward.cpu(), ts_c.reward, rtol=2e-3, atol=2e-3)
        torch.testing.assert_close(ts_g.step_type.cpu(), ts_c.step_type)
        if bool(ts_c.last().any()):
            gpu_env._state["s"].copy_(cpu_env._state["s"].cuda())
        else:
            # keep fp drift bounded over the comparison horizon
            gpu_env._state["s"].copy_(cpu_env._state["s"].cuda())


@requires_gpu
def test_ant_episode_metrics_gpu(ext):
    from stoix_amd.envs.ant import Ant

    env = Ant(128, device="cuda", seed=0)
    env.max_episode_steps = 10
    env.reset()
    for _ in range(10):
        ts = env.step(torch.zeros(128, 8, device="cuda"))
    em = ts.extras["episode_metrics"]
    done = em["is_terminal_step"]
    assert bool(done.all())  # truncation fires for every env at step 10
    assert (em["episode_length"][done] == 10).all()


@requires_gpu
def test_ppo_gpu_update_step(ext):
    """Full PPO update step on GPU with HIP env + GAE kernels."""
    from stoix_amd.config import compose
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd import envs as environments
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        [
            "env=brax/ant",
            "arch.total_num_envs=512",
            "arch.total_timesteps=null",
            "arch.num_updates=4",
            "arch.num_evaluation=1",
            "system.rollout_length=8",
            "system.num_minibatches=2",
            "system.epochs=1",
            "system.compute_dtype=bf16",
            "logger.loggers=[]",
        ],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    env = environments.make_single(cfg, 512, device, seed=0)
    learner = PPOLearner(cfg, env, device)
    for _ in range(3):
        m = learner.update_step()
    torch.cuda.synchronize()
    for k, v in m.items():
        assert torch.isfinite(v), k


@requires_gpu
def test_ppo_graph_capture(ext):
    """hip-graph capture of the whole update step replays correctly."""
    from stoix_amd.config import compose
    from stoix_amd.ops.graph import try_enable_graphs
    from stoix_amd.systems.ppo.ff_ppo import PPOLearner
    from stoix_amd import envs as environments
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    cfg = compose(
        "default/anakin/default_ff_ppo_continuous.yaml",
        [
            "env=brax/ant",
            "arch.total_num_envs=512",
            "arch.total_timesteps=null",
            "arch.num_updates=4",
            "arch.num_evaluation=1",
            "system.rollout_length=8",
            "system.num_minibatches=2",
            "system.epochs=1",
            "system.compute_dtype=bf16",
            "logger.loggers=[]",
        ],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    device = torch.device("cuda:0")
    env = environments.make_single(cfg, 512, device, seed=0)
    learner = PPOLearner(cfg, env, device)
    ok = try_enable_graphs(learner)
    assert ok
    for _ in range(3):
        m = learner.update_step()
    torch.cuda.synchronize()
    for k, v in m.items():
        assert torch.isfinite(v), k
    # params actually move under replay
    p = next(learner.actor.parameters())
    before = p.detach().clone()
    for _ in range(3):
        learner.update_step()
    torch.cuda.synchronize()
    assert not torch.equal(before, p.detach())


@requires_gpu
def test_humanoid_kernel_matches_torch_env(ext):
    """HIP humanoid_step vs the torch reference physics for identical
    states/actions (single non-terminal steps)."""
    from stoix_amd.envs.humanoid import Humanoid

    torch.manual_seed(0)
    B = 64
    env_g = Humanoid(num_envs=B, device="cuda", seed=5)
    ts = env_g.reset()
    env_c = Humanoid(num_envs=B, device="cpu", seed=5)
    env_c.reset()
    g = torch.Generator().manual_seed(1)
    for i in range(5):
        # keep CPU state mirror synchronised with the GPU state
        env_c._state = {"s": env_g._state["s"].detach().cpu().clone()}
        a = (torch.rand(B, 17, generator=g) * 2 - 1)
        state_c, reward_c, term_c = env_c._step_fn({"s": env_c._state["s"]}, a)
        ts = env_g.step(a.cuda())
        torch.cuda.synchronize()
        # compare PRE-autoreset next_obs against the CPU physics
        obs_c = env_c._obs_fn(state_c)
        torch.testing.assert_close(
            ts.extras["next_obs"].cpu(), obs_c, rtol=2e-4, atol=2e-4
        )
        torch.testing.assert_close(ts.reward.cpu(), reward_c, rtol=2e-4, atol=2e-3)


@requires_gpu
def test_hip_env_step_outputs_are_stable_across_steps(ext):
    """Regression (GPU-env aliasing): the TimeStep returned by step t must
    stay unchanged after step t+1 runs — learners hold these tensors
    across subsequent env steps (some for the whole rollout)."""
    from stoix_amd.envs.classic import CartPole

    env = CartPole(num_envs=32, device="cuda:0", seed=0)
    env.reset()
    a = torch.zeros(32, dtype=torch.long, device="cuda:0")
    ts1 = env.step(a)
    obs1 = ts1.observation.clone()
    r1 = ts1.reward.clone()
    n1 = ts1.extras["next_obs"].clone()
    for _ in range(3):
        env.step(a)
    torch.cuda.synchronize()
    torch.testing.assert_close(ts1.observation, obs1, rtol=0, atol=0)
    torch.testing.assert_close(ts1.reward, r1, rtol=0, atol=0)
    torch.testing.assert_close(ts1.extras["next_obs"], n1, rtol=0, atol=0)


# ------------------------------------------------------------- sum-tree PER


@requires_gpu
def test_sumtree_hip_update_matches_cpu_reference(ext):
    """HIP sumtree_update (scatter + per-level ancestor repair) vs the
    torch fallback on CPU, including the multi-launch path (n > 4096) and
    duplicate indices."""
    from stoix_amd.buffers.per import SumTree

    for n_items, n_upd in [(1000, 256), (300000, 9000)]:
        cpu = SumTree(n_items, "cpu")
        gpu = SumTree(n_items, "cuda")
        g = torch.Generator().manual_seed(7)
        # UNIQUE base indices (duplicate indices with different priorities
        # have an intentionally arbitrary scatter winner), plus controlled
        # duplicates with identical priorities (the defined case)
        idx = torch.randperm(n_items, generator=g)[:n_upd].contiguous()
        prio = torch.rand(n_upd, generator=g) + 0.01
        idx[: n_upd // 4] = idx[n_upd // 4 : n_upd // 2]
        prio[: n_upd // 4] = prio[n_upd // 4 : n_upd // 2]
        cpu.set(idx, prio)
        gpu.set(idx.cuda(), prio.cuda())
        torch.testing.assert_close(gpu.tree.cpu(), cpu.tree, rtol=1e-5, atol=1e-5)
        assert abs(float(gpu.total) - float(cpu.total)) < 1e-3


@requires_gpu
def test_sumtree_hip_sample_proportional(ext):
    """HIP stratified descent samples proportionally to priorities."""
    from stoix_amd.buffers.per import SumTree

    t = SumTree(64, "cuda")
    idx = torch.arange(64, device="cuda")
    prio = torch.zeros(64, device="cuda")
    prio[3] = 1.0
    prio[40] = 3.0
    t.set(idx, prio)
    gen = torch.Generator(device="cuda").manual_seed(3)
    s = t.sample(4000, gen)
    frac40 = float((s == 40).float().mean())
    frac3 = float((s == 3).float().mean())
    assert abs(frac40 - 0.75) < 0.03
    assert abs(frac3 - 0.25) < 0.03
    assert set(s.unique().cpu().tolist()) <= {3, 40}


@requires_gpu
def test_rainbow_update_graph_captured_on_snake():
    """The whole Rainbow update (rollout + PER sample + loss + priority
    writeback + polyak) captures into ONE hip graph on the capture-safe
    Snake env, replays without host syncs, and the tree stays consistent."""
    import importlib

    from stoix_amd import envs as environments
    from stoix_amd.config import compose
    from stoix_amd.ops.graph import try_enable_update_graph
    from stoix_amd.utils.total_timestep_checker import check_total_timesteps

    mod = importlib.import_module("stoix_amd.systems.q_learning.ff_rainbow")
    cfg = compose(
        "default/anakin/default_ff_rainbow.yaml",
        ["env=jumanji/snake", "arch.total_num_envs=64",
         "arch.total_timesteps=null", "arch.num_updates=8",
         "arch.num_evaluation=1", "system.rollout_length=4",
         "system.batch_size=64", "system.buffer_size=4096",
         "system.warmup_steps=16", "system.n_step=3", "system.epochs=2",
         "logger.loggers=[]"],
    )
    cfg.arch.n_devices = 1
    check_total_timesteps(cfg)
    dev = torch.device("cuda:0")
    env = environments.make_single(cfg, 64, dev, seed=0)
    learner = mod.RainbowLearner(cfg, env, dev)
    assert learner.graph_capturable
    ok = try_enable_update_graph(learner)
    assert ok and getattr(learner, "_graphs", None) is not None
    for _ in range(4):
        m = learner.update_step()
    torch.cuda.synchronize()
    assert torch.isfinite(m["q_loss"]).all()
    # tree invariant: root equals the sum of the leaves after replays
    tree = learner.buffer.tree
    leaf_sum = float(tree.tree[tree.capacity :].sum())
    assert abs(float(tree.total) - leaf_sum) / max(leaf_sum, 1.0) < 1e-3
    # priorities were written back (not all still at the max-fill value)
    leaves = tree.tree[tree.capacity : tree.capacity + tree.n_items]
    nz = leaves[leaves > 0]
    assert nz.numel() > 0 and float(nz.std()) > 1e-6


# ------------------------------------------------------------ K13 RNN scan


@requires_gpu
@pytest.mark.parametrize("kind,H", [("gru", 128), ("gru", 256), ("lstm", 128), ("lstm", 256)])
def test_rnn_scan_kernel_matches_eager(ext, kind, H):
    """Fused done-masked RNN scan (ops/csrc/rnn.hip) vs the fp32 per-step
    reference, with random resets; bf16-GEMM-class tolerance over T=40."""
    from stoix_amd.networks.base import ScannedRNN

    torch.manual_seed(0)
    rnn = ScannedRNN(37, H, cell_type=kind).cuda()
    T, B = 40, 133
    x = torch.randn(T, B, 37, device="cuda")
    resets = torch.rand(T, B, device="cuda") < 0.15
    st0 = rnn.initial_state(B, "cuda")
    with torch.no_grad():
        out_hip, st_hip = rnn(x, resets, list(st0))  # dispatches to rnn_scan
    orig = rnn._single_cell
    rnn._single_cell = lambda: (None, None)
    with torch.no_grad():
        out_ref, st_ref = rnn(x, resets, list(st0))
    rnn._single_cell = orig
    torch.testing.assert_close(out_hip, out_ref, rtol=5e-2, atol=3e-2)
    if kind == "gru":
        torch.testing.assert_close(st_hip[0], st_ref[0], rtol=5e-2, atol=3e-2)
    else:
        torch.testing.assert_close(st_hip[0][0], st_ref[0][0], rtol=5e-2, atol=3e-2)
        torch.testing.assert_close(st_hip[0][1], st_ref[0][1], rtol=5e-2, atol=5e-2)


@requires_gpu
def test_snake_kernel_matches_torch_env(ext):
    """Fused Snake step (ops/csrc/snake.hip) vs the tensorised torch path
    on identical states; resync after eat/reset events (RNG streams for
    fruit/reset placement differ by construction, like the CartPole
    reset-noise protocol)."""
    from stoix_amd.envs.snake import Snake

    cpu_env = Snake(32, device="cpu", seed=0)
    gpu_env = Snake(32, device="cuda", seed=0)
    assert gpu_env._hip is not None
    cpu_env.reset()
    gpu_env.reset()

    def sync():
        for k in cpu_env._state:
            gpu_env._state[k].copy_(cpu_env._state[k].cuda())

    sync()
    g = torch.Generator().manual_seed(4)
    for i in range(60):
        a = torch.randint(0, 4, (32,), generator=g)
        ts_c = cpu_env.step(a)
        ts_g = gpu_env.step(a.cuda())
        torch.testing.assert_close(ts_g.reward.cpu(), ts_c.reward)
        torch.testing.assert_close(ts_g.discount.cpu(), ts_c.discount)
        torch.testing.assert_close(ts_g.step_type.cpu(), ts_c.step_type)
        # true-final obs parity except envs that ATE (fruit respawn RNG
        # differs -> fruit channel differs there)
        calm = (ts_c.reward == 0) & ~ts_c.extras["episode_metrics"]["is_terminal_step"]
        torch.testing.assert_close(
            ts_g.extras["next_obs"].cpu()[calm], ts_c.extras["next_obs"][calm]
        )
        torch.testing.assert_close(
            ts_g.observation.cpu()[calm], ts_c.observation[calm]
        )
        em_c = ts_c.extras["episode_metrics"]
        em_g = ts_g.extras["episode_metrics"]
        torch.testing.assert_close(em_g["episode_return"].cpu(), em_c["episode_return"])
        torch.testing.assert_close(em_g["episode_length"].cpu(), em_c["episode_length"])
        sync()
        cpu_env._step_count.copy_(cpu_env._step_count)  # no-op, clarity
        gpu_env._step_count.copy_(cpu_env._step_count.cuda())
        gpu_env._ep_return.copy_(cpu_env._ep_return.cuda())
        gpu_env._ep_length.copy_(cpu_env._ep_length.cuda())
