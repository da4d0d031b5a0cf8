#!/usr/bin/env python3
"""A/B the payload-permute variants at bench scale: span gather (3, the
default), nt gather (4), and the sequential-read/random-write scatter
probe (5). Uses a uniform random permutation (the post-sort index
distribution of uniform keys)."""
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G           # noqa: E402
from thrill_amd import Native         # noqa: E402

REC = 100
N = int(os.environ.get("T9_AB_N", 10 * 1024**3 // REC))

nat = Native(device=0)
din = G.empty(N * REC, np.uint8)
dout = G.empty(N * REC, np.uint8)
nat.gen_records(G.ptr(din), 0, N, 1, G.stream())
perm = np.random.default_rng(5).permutation(N).astype(np.uint32)
didx = G.dev(perm)
res = {"records": N}
for var in [3, 4, 5]:
    os.environ["T9_GATHER_VARIANT"] = str(var)
    s = G.stream()
    nat.gather_records(G.ptr(din), G.ptr(didx), N, REC, G.ptr(dout), s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        nat.gather_records(G.ptr(din), G.ptr(didx), N, REC, G.ptr(dout), s)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 3 * 1e3
    res[f"var{var}_ms"] = round(ms, 3)
    res[f"var{var}_algo_TBps"] = round(204.0 * N / (ms * 1e-3) / 1e12, 2)
del os.environ["T9_GATHER_VARIANT"]
print(json.dumps(res), flush=True)
nat.close()
