"""Micro-bench for the stacked actor+critic epoch redesign.

Times, at the fused-PPO minibatch shape (S=32768, H=256, K1P=32):
  a) current: 2x linear_silu per layer (actor + critic separately)
  b) stacked: torch.baddbmm over [2, S, *] batches (stride-0 expanded X for
     layer 1, transposed weight views) + flat silu_fwd
plus the building blocks, to confirm hipBLASLt consumes the strided views
directly (no silent .contiguous() copies).
Run on a GPU box: python tools/bmm_probe.py
"""
from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from stoix_amd import ops

S, H, K1P = 32768, 256, 32


def step(msg):
    torch.cuda.synchronize()
    print("::", msg, flush=True)


def timeit(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    assert torch.cuda.is_available()
    ext = ops.ext(required=True)
    dev = torch.device("cuda:0")
    # enable the shipped tunableop selections like the engine does
    tun = torch.cuda.tunable
    tun.enable(True)
    tun.tuning_enable(False)
    tun.read_file(os.path.join(os.path.dirname(__file__), "..",
                               "stoix_amd", "ops", "tunableop_gfx950.csv"))

    bf = torch.bfloat16
    X = torch.randn(S, K1P, device=dev, dtype=bf)
    W1 = torch.randn(2, H, K1P, device=dev, dtype=bf)  # [net, out, in] contiguous
    b1 = torch.randn(2, 1, H, device=dev, dtype=bf)
    W2 = torch.randn(2, H, H, device=dev, dtype=bf)
    b2 = torch.randn(2, 1, H, device=dev, dtype=bf)
    Z1 = torch.zeros(2, S, H, device=dev, dtype=bf)
    H1 = torch.zeros(2, S, H, device=dev, dtype=bf)
    Z2 = torch.zeros(2, S, H, device=dev, dtype=bf)
    H2 = torch.zeros(2, S, H, device=dev, dtype=bf)

    # single-net buffers for the current path
    W1s = W1[0].contiguous()
    b1s = torch.randn(H, device=dev, dtype=torch.float32)
    W2s = W2[0].contiguous()
    Z1s = torch.zeros(S, H, device=dev, dtype=bf)
    H1s = torch.zeros(S, H, device=dev, dtype=bf)

    Xe = X.unsqueeze(0).expand(2, S, K1P)  # stride-0 batch
    W1t = W1.transpose(1, 2)  # [2, K1P, H] T-view
    W2t = W2.transpose(1, 2)

    print("expanded X strides:", Xe.stride(), "W1t strides:", W1t.stride(), flush=True)

    step("linear_silu L1")
    t = timeit(lambda: ext.linear_silu(X, W1s, b1s, Z1s, H1s, 1))
    print(f"linear_silu L1 (one net)      {t:8.2f} us  -> x2 = {2*t:.2f}")
    step("linear_silu L2")
    H1in = H1s.clone()
    t = timeit(lambda: ext.linear_silu(H1in, W2s, b1s, Z1s, H1s, 1))
    print(f"linear_silu L2 (one net)      {t:8.2f} us  -> x2 = {2*t:.2f}")

    step("baddbmm L1")
    t = timeit(lambda: torch.baddbmm(b1, Xe, W1t, out=Z1))
    print(f"baddbmm L1 stacked (stride0)  {t:8.2f} us")
    step("baddbmm L2")
    t = timeit(lambda: torch.baddbmm(b2, H1, W2t, out=Z2))
    print(f"baddbmm L2 stacked            {t:8.2f} us")
    step("silu_fwd")
    t = timeit(lambda: ext.silu_fwd(Z1, H1))
    print(f"silu_fwd stacked (2S*H)       {t:8.2f} us")

    # backward-side shapes: dH = dZ @ W (NN), stacked as bmm
    dZ = torch.randn(2, S, H, device=dev, dtype=bf)
    dH = torch.zeros(2, S, H, device=dev, dtype=bf)
    step("mm dH")
    t = timeit(lambda: torch.mm(dZ[0], W2[0], out=dH[0]))
    print(f"mm dH (one net)               {t:8.2f} us  -> x2 = {2*t:.2f}")
    step("bmm dH")
    t = timeit(lambda: torch.bmm(dZ, W2, out=dH))
    print(f"bmm dH stacked                {t:8.2f} us")

    # memory sanity: no hidden copies -> allocator stats stable
    torch.cuda.reset_peak_memory_stats()
    for _ in range(5):
        torch.baddbmm(b1, Xe, W1t, out=Z1)
        torch.baddbmm(b2, H1, W2t, out=Z2)
    torch.cuda.synchronize()
    print("peak alloc during bmms (MB):",
          torch.cuda.max_memory_allocated() / 1e6)


if __name__ == "__main__":
    main()
