import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from stoix_amd.envs.classic import CartPole

def trunc_stats(hip: bool, steps=400):
    torch.manual_seed(0)
    env = CartPole(num_envs=64, device="cuda:0", seed=0)
    env.max_episode_steps = 25  # force truncations with a do-nothing-ish policy
    if not hip:
        env._hip = None
    ts = env.reset()
    g = torch.Generator(device="cuda:0"); g.manual_seed(1)
    n_term = n_trunc = 0
    lens = []
    viol = {}
    for i in range(steps):
        # alternating actions keep the pole up longer -> more truncations
        a = torch.full((64,), i % 2, device="cuda:0", dtype=torch.long)
        ts = env.step(a)
        st = ts.step_type
        n_term += int((st == 2).sum())
        n_trunc += int((st == 3).sum())
        done = ts.extras["episode_metrics"]["is_terminal_step"]
        if bool(done.any()):
            lens += ts.extras["episode_metrics"]["episode_length"][done].tolist()
        # at truncation: discount must be 1, and next_obs must be valid
        tr = st == 3
        if tr.any():
            if not bool((ts.discount[tr] == 1).all()):
                viol["disc_at_trunc"] = viol.get("disc_at_trunc", 0) + 1
            if bool(done[tr].logical_not().any()):
                viol["done_at_trunc"] = viol.get("done_at_trunc", 0) + 1
        mx = max(lens) if lens else 0
    import statistics
    print(f"hip={hip}: term={n_term} trunc={n_trunc} max_len={mx} "
          f"mean_len={statistics.mean(lens):.1f} viol={viol}")

trunc_stats(False)
trunc_stats(True)
