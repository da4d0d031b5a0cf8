#!/usr/bin/env python3
"""A/B the 128-bit reduce build LDS filter size (T9_LDS128_SLOTS) and
grid on a config-4-shaped stream (Zipf 1.1, 10M vocab, 2^29 tokens)."""
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G                     # noqa: E402
from thrill_amd import Native                   # noqa: E402
from thrill_amd.pipeline import zipf_cdf        # noqa: E402

n = 1 << 29
vocab = 10_000_000
nat = Native(device=0)
cdf = torch.from_numpy(zipf_cdf(vocab, 1.1)).cuda()
toks = G.empty(n, np.uint64)
nat.zipf_tokens(G.ptr(toks), G.ptr(cdf), vocab, 0, n, 0x44, G.stream())
k1, k2 = G.empty(n, np.uint64), G.empty(n, np.uint64)
nat.hash2_of(G.ptr(toks), n, G.ptr(k1), G.ptr(k2), G.stream())
cap = 1 << 25
tbl = G.empty(4 * cap, np.uint64)   # sized for the stride-4 variant too
derr = G.empty(1, np.uint32)
res = {"n": n, "vocab": vocab}
for slots, rf, sstr in [(4096, 0, 3), (4096, 1, 3), (4096, 1, 4)]:
    os.environ["T9_LDS128_SLOTS"] = str(slots)
    os.environ["T9_R128_READFIRST"] = str(rf)
    os.environ["T9_R128_STRIDE"] = str(sstr)
    s = G.stream()

    def one():
        nat.reduce128_init(G.ptr(tbl), cap, s)
        nat.reduce128_build(G.ptr(k1), G.ptr(k2), None, n, G.ptr(tbl),
                            cap, 0, G.ptr(derr), s)
    one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        one()
    torch.cuda.synchronize()
    res[f"slots{slots}_rf{rf}_s{sstr}_ms"] = round(
        (time.perf_counter() - t0) / 3 * 1e3, 2)
del os.environ["T9_LDS128_SLOTS"]
del os.environ["T9_R128_READFIRST"]
del os.environ["T9_R128_STRIDE"]
print(json.dumps(res), flush=True)
nat.close()
