#!/usr/bin/env python3
"""Time LSD vs MSB pair sort on 2^27 uniform pairs."""
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G          # noqa: E402
from thrill_amd import Native        # noqa: E402

nat = Native(device=0)
n = 1 << 27
s = G.stream()
src_k = G.empty(n, np.uint64)
nat.gen_u64(G.ptr(src_k), 0, n, 0x7421, s)
src_v = torch.arange(n, dtype=torch.int32, device="cuda")
dk = torch.empty_like(src_k)
dv = torch.empty_like(src_v)
w = G.ws(nat.ws("sort_pairs", n))
torch.cuda.synchronize()

for algo in ["lsd", "msb"]:
    os.environ["T9_SORT_ALGO"] = algo

    def run():
        dk.copy_(src_k)
        dv.copy_(src_v)
        nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), s)

    run()
    torch.cuda.synchronize()
    signed = dk ^ (-2 ** 63)
    assert bool((signed[1:] >= signed[:-1]).all().item()), algo
    assert int(dk.sum().item()) == int(src_k.sum().item()), algo
    ts = []
    for _ in range(3):
        t0 = time.perf_counter()
        run()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = min(ts)
    print(json.dumps({algo: {"sort_s": round(t, 4),
                             "Mpairs_per_s": round(n / t / 1e6, 1)}}),
          flush=True)
wk = G.ws(nat.ws("sort_u64", n))
for algo in ["lsd", "msb"]:
    os.environ["T9_SORT_ALGO"] = algo

    def runk():
        dk.copy_(src_k)
        nat.sort_u64(G.ptr(dk), n, G.ptr(wk), s)

    runk()
    torch.cuda.synchronize()
    signed = dk ^ (-2 ** 63)
    assert bool((signed[1:] >= signed[:-1]).all().item()), algo
    ts = []
    for _ in range(3):
        t0 = time.perf_counter()
        runk()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = min(ts)
    print(json.dumps({f"keys_{algo}": {"sort_s": round(t, 4),
                      "Mkeys_per_s": round(n / t / 1e6, 1)}}), flush=True)
nat.close()
