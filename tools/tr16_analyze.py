"""Decode __builtin_amdgcn_ds_read_tr16_b64_v4bf16 semantics empirically.

Stages lds[i] = i (0..1023) and dumps, for each base mode (0: uniform base,
1: per-lane base + (l>>4)*64 elems, 2: + (l>>4)*128), which source element
each (lane, j) received. Prints the inferred index formula residuals for a
few candidate mappings.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from stoix_amd import ops

ext = ops.ext(required=True)
inp = torch.arange(1024, dtype=torch.float32).bfloat16().cuda()
for mode in (0, 1, 2):
    out = torch.zeros(64 * 4, device="cuda")
    ext.tr16_probe(inp, out, mode)
    torch.cuda.synchronize()
    got = out.view(64, 4).long().cpu()
    print(f"--- mode {mode}: lane -> 4 source elements")
    for l in [0, 1, 2, 3, 15, 16, 17, 31, 32, 48, 63]:
        print(f"  lane {l:2d}: {got[l].tolist()}")
    # candidate formulas
    import itertools
    cands = {
        "guide: (l&15)+j*16+(l>>4)*64": lambda l, j: (l & 15) + j * 16 + (l >> 4) * 64,
        "rowmajor16: (l&15)+j*16": lambda l, j: (l & 15) + j * 16,
        "linear: l*4+j": lambda l, j: l * 4 + j,
        "quad: (l>>4)*64+(l&15)*4+j": lambda l, j: (l >> 4) * 64 + (l & 15) * 4 + j,
    }
    base = {0: lambda l: 0, 1: lambda l: (l >> 4) * 64, 2: lambda l: (l >> 4) * 128}[mode]
    for name, f in cands.items():
        ok = all(int(got[l, j]) == f(l, j) + base(l) for l in range(64) for j in range(4))
        if ok:
            print(f"  MATCH({mode}): base + {name}")
