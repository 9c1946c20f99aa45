"""Tune the strided-batched dH1 GEMM shape with TunableOp and emit the
resulting CSV lines (merge into stoix_amd/ops/tunableop_gfx950.csv).
Run on a GPU box: python tools/tune_bmm.py ; results land in
gpurun_out/tunableop_bmm.csv
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "200")
import torch

S, H = 32768, 256


def main():
    tun = torch.cuda.tunable
    tun.enable(True)
    tun.tuning_enable(True)
    csv = os.path.join(os.path.dirname(__file__), "..", "stoix_amd", "ops",
                       "tunableop_gfx950.csv")
    if os.path.exists(csv):
        tun.read_file(csv)
    dev = torch.device("cuda:0")
    bf = torch.bfloat16
    dZ2 = torch.randn(2, S, H, device=dev, dtype=bf)
    W2pair = torch.randn(2, H, H, device=dev, dtype=bf)
    dH1 = torch.zeros(2, S, H, device=dev, dtype=bf)
    for _ in range(20):
        torch.bmm(dZ2, W2pair, out=dH1)
    torch.cuda.synchronize()
    os.makedirs("gpurun_out", exist_ok=True)
    tun.write_file("gpurun_out/tunableop_bmm.csv")
    import time

    t0 = time.perf_counter()
    for _ in range(200):
        torch.bmm(dZ2, W2pair, out=dH1)
    torch.cuda.synchronize()
    print(f"tuned bmm: {(time.perf_counter()-t0)/200*1e6:.2f} us")
    with open("gpurun_out/tunableop_bmm.csv") as f:
        for line in f:
            if "Batched" in line or "bmm" in line.lower():
                print("CSV:", line.strip())


if __name__ == "__main__":
    main()
