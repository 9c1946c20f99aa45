import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from stoix_amd.envs.classic import CartPole

def stats(hip: bool, steps=1000):
    torch.manual_seed(0)
    env = CartPole(num_envs=128, device="cuda:0", seed=0)
    if not hip:
        env._hip = None
    ts = env.reset()
    g = torch.Generator(device="cuda:0"); g.manual_seed(1)
    lens, rets = [], []
    viol = 0
    for i in range(steps):
        a = torch.randint(0, 2, (128,), device="cuda:0", generator=g)
        ts = env.step(a)
        done = ts.extras["episode_metrics"]["is_terminal_step"]
        if bool(done.any()):
            em = ts.extras["episode_metrics"]
            lens += em["episode_length"][done].tolist()
            rets += em["episode_return"][done].tolist()
            # post-reset obs must be in [-0.05, 0.05]
            ro = ts.observation[done]
            if (ro.abs() > 0.05001).any(): viol += 1
        # invariant: where not done, observation == next_obs
        nd = ~done
        if nd.any():
            d = (ts.observation[nd] - ts.extras["next_obs"][nd]).abs().max().item()
            if d > 0: viol += 100
        # discount/steptype consistency
        if not bool(((ts.discount == 0) == (ts.step_type == 2)).all()): viol += 10000
    import statistics
    print(f"hip={hip}: n_eps={len(lens)} mean_len={statistics.mean(lens):.1f} "
          f"mean_ret={statistics.mean(rets):.1f} viol={viol}")

stats(False)
stats(True)
