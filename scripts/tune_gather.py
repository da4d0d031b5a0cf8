#!/usr/bin/env python3
"""Gather variant timing at bench scale (1.07e8 records, random perm)."""
import json, os, sys, time
import numpy as np, torch
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G
from thrill_amd import Native

nat = Native(device=0)
n = 107_374_182
s = G.stream()
din = G.empty(n * 100, np.uint8)
nat.gen_records(G.ptr(din), 0, n, 1, s)
didx = torch.randperm(n, device="cuda").to(torch.int32)
dout = G.empty(n * 100, np.uint8)
torch.cuda.synchronize()
for var in [1, 2, 3, 4]:
    for grid in ([4096, 8192, 16384] if var in (1, 3) else [4096]):
        os.environ["T9_GATHER_VARIANT"] = str(var)
        os.environ["T9_GATHER_GRID"] = str(grid)
        nat.gather_records(G.ptr(din), G.ptr(didx), n, 100, G.ptr(dout), s)
        torch.cuda.synchronize()
        ts = []
        for _ in range(3):
            t0 = time.perf_counter()
            nat.gather_records(G.ptr(din), G.ptr(didx), n, 100,
                               G.ptr(dout), s)
            torch.cuda.synchronize()
            ts.append(time.perf_counter() - t0)
        print(json.dumps({f"v{var}_g{grid}": round(min(ts) * 1e3, 2)}),
              flush=True)
# correctness spot check on small n with variant 3
os.environ["T9_GATHER_VARIANT"] = "3"
os.environ["T9_GATHER_GRID"] = "4096"
m = 100_000
idx = torch.randperm(m, device="cuda").to(torch.int32)
o2 = G.empty(m * 100, np.uint8)
nat.gather_records(G.ptr(din), G.ptr(idx), m, 100, G.ptr(o2), s)
got = G.host(o2, np.uint8).reshape(m, 100)
src = G.host(din, np.uint8)[:m * 100].reshape(m, 100)
assert np.array_equal(got, src[G.host(idx, np.uint32).astype(np.int64)])
print("span variant correct")
nat.close()
