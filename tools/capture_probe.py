"""Diagnose hip-graph capture legality of individual torch ops at bench
sizes. Run on a GPU box; prints one PASS/FAIL line per op."""
import sys

import torch


def probe(name, fn, warmup=2):
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"PASS {name}")
    except Exception as e:
        print(f"FAIL {name}: {type(e).__name__}: {str(e)[:150]}")


def main():
    dev = torch.device("cuda")
    TB = 128 * 4096

    x = torch.rand(TB, device=dev)
    probe("rand_524288", lambda: torch.rand(TB, device=dev))
    probe("argsort_524288", lambda: torch.argsort(torch.rand(TB, device=dev)))
    probe("argsort_4096", lambda: torch.argsort(torch.rand(4096, device=dev)))
    probe("randperm_524288", lambda: torch.randperm(TB, device=dev))

    lin = torch.nn.Sequential(
        torch.nn.Linear(27, 256), torch.nn.SiLU(), torch.nn.Linear(256, 256), torch.nn.SiLU(), torch.nn.Linear(256, 8)
    ).cuda()
    inp = torch.randn(32768, 27, device=dev)

    def fwd_bwd():
        with torch.autocast("cuda", torch.bfloat16):
            out = lin(inp).float().square().mean()
        out.backward()

    probe("mlp_fwd_bwd_bf16_32768", fwd_bwd)

    opt = torch.optim.Adam(lin.parameters(), capturable=True)

    def clip_step():
        torch.nn.utils.clip_grad_norm_(lin.parameters(), 0.5)
        opt.step()
        opt.zero_grad(set_to_none=False)

    for p in lin.parameters():
        p.grad = torch.zeros_like(p)
    probe("clip_adam_capturable", clip_step)

    gather = torch.randn(TB, 27, device=dev)
    idx = torch.randint(0, TB, (32768,), device=dev)
    probe("gather_rows", lambda: gather[idx])

    probe("std_mean", lambda: (x - x.mean()) / (x.std() + 1e-8))


if __name__ == "__main__":
    main()
