#!/usr/bin/env python3
"""Establish the gather kernel's memory-wall context: identical kernel,
three index distributions (sequential / window-local random / global
random) at bench scale."""
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
dout = G.empty(n * 100, np.uint8)

def timeit(didx):
    nat.gather_records(G.ptr(din), G.ptr(didx), n, 100, G.ptr(dout), s)
    torch.cuda.synchronize()
    ts = []
    for _ in range(3):
        t0 = time.perf_counter()
        nat.gather_records(G.ptr(din), G.ptr(didx), n, 100, G.ptr(dout), s)
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    return round(min(ts) * 1e3, 2)

seq = torch.arange(n, dtype=torch.int32, device="cuda")
print(json.dumps({"sequential_ms": timeit(seq)}), flush=True)
# window-local random: shuffle within 16Ki-record (1.6 MB) windows
W = 1 << 14
nw = n // W
base = torch.arange(nw, device="cuda").repeat_interleave(W) * W
offs = torch.argsort(torch.rand(nw, W, device="cuda"), dim=1).reshape(-1)
local = (base + offs).to(torch.int32)
pad = torch.arange(nw * W, n, dtype=torch.int32, device="cuda")
local = torch.cat([local, pad])
print(json.dumps({"window16k_random_ms": timeit(local)}), flush=True)
rnd = torch.randperm(n, device="cuda").to(torch.int32)
print(json.dumps({"global_random_ms": timeit(rnd)}), flush=True)
nat.close()
