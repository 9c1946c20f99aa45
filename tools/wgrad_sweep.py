"""Sweep wgrad (KPG, NTB) per flagship shape on the GPU and print the
fastest variant for each (bake the winners into wgrad.hip's launcher).
Shapes (S=32768): W2 N=256 K=256; W1 N=256 K=32; Wh N=16 K=256; Wv N=1 K=256.
"""
from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from stoix_amd import ops

S = 32768


def timeit(fn, iters=100, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ext = ops.ext(required=True)
    dev = torch.device("cuda:0")
    bf = torch.bfloat16
    shapes = [("W2", 256, 256, 256), ("W1", 256, 32, 256), ("Wh", 16, 256, 16), ("Wv", 1, 256, 16)]
    for name, NV, K, NSTRIDE in shapes:
        dZ = torch.randn(S, NSTRIDE, device=dev, dtype=bf)
        X = torch.randn(S, K, device=dev, dtype=bf)
        # realistic slab: the engine's chain slab has stride ~74.5K floats
        # and this dW sits at a mid-chain offset
        stride = 74512
        off = 8448
        slab = torch.zeros(64, stride, device=dev, dtype=torch.float32)
        NT = (NV + 15) // 16
        results = []
        for kpg in (1, 2, 4, 8):
            if K % (16 * kpg) != 0:
                continue
            for ntb in (1, 4):
                if NT % ntb != 0:
                    continue
                wgs = (NT // ntb) * (K // (16 * kpg)) * 16
                t = timeit(lambda: ext.wgrad(dZ, X, slab, off, off + NV * K, NV, kpg, ntb))
                results.append((t, kpg, ntb, wgs))
        results.sort()
        best = results[0]
        print(f"{name} (N={NV},K={K}): best KPG={best[1]} NTB={best[2]} "
              f"({best[3]} WGs) {best[0]:.2f} us | " +
              " ".join(f"k{k}n{n}={t:.1f}({w}wg)" for t, k, n, w in results))


if __name__ == "__main__":
    main()
