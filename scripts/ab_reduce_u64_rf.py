#!/usr/bin/env python3
"""A/B read-before-CAS on the u64 reduce build (T9_REDUCE_READFIRST)."""
import json, os, sys, time
import numpy as np, torch
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G
from thrill_amd import Native
from thrill_amd.pipeline import zipf_cdf

n = 1 << 29
vocab = 10_000_000
nat = Native(device=0)
cdf = torch.from_numpy(zipf_cdf(vocab, 1.1)).cuda()
toks = G.empty(n, np.uint64)
nat.zipf_tokens(G.ptr(toks), G.ptr(cdf), vocab, 0, n, 0x44, G.stream())
ones = torch.ones(n, dtype=torch.int64, device="cuda")
cap = 1 << 25
tbl = G.empty(2 * (cap + 1), np.uint64)
derr = G.empty(1, np.uint32)
res = {"n": n, "vocab": vocab}
for rf in [0, 1]:
    os.environ["T9_REDUCE_READFIRST"] = str(rf)
    s = G.stream()
    def one():
        nat.reduce_init(G.ptr(tbl), cap, s)
        nat.reduce_build(G.ptr(toks), G.ptr(ones), n, G.ptr(tbl), cap, 0,
                         G.ptr(derr), s)
    one(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        one()
    torch.cuda.synchronize()
    res[f"rf{rf}_ms"] = round((time.perf_counter() - t0) / 3 * 1e3, 2)
del os.environ["T9_REDUCE_READFIRST"]
print(json.dumps(res), flush=True)
nat.close()
