#!/usr/bin/env python3
"""A/B the fused extract buffering (double vs single LDS buffer) at
bench scale (occupancy 3 vs 6 blocks/CU; the wave-cycle decomposition
shows the kernel parked on the stage barrier)."""
import json, os, sys, time
import numpy as np, torch
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G
from thrill_amd import Native

REC = 100
N = 10 * 1024**3 // REC
nat = Native(device=0)
din = G.empty(N * REC, np.uint8)
dout = G.empty(N * REC, np.uint8)
w = G.ws(nat.ws("sort_records", N, REC))
nat.gen_records(G.ptr(din), 0, N, 1, G.stream())
res = {}
for sb in [0, 1]:
    os.environ["T9_EXTRACT_SB"] = str(sb)
    s = G.stream()
    nat.sort_records(G.ptr(din), G.ptr(dout), N, REC, 10, G.ptr(w), s)
    torch.cuda.synchronize()
    nat.perf_reset(); nat.perf_enable(True)
    t0 = time.perf_counter()
    for _ in range(3):
        nat.sort_records(G.ptr(din), G.ptr(dout), N, REC, 10, G.ptr(w), s)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    nat.perf_enable(False)
    ms, cnt = nat.perf_read("extract")
    res[f"sb{sb}_step_ms"] = round(dt * 1e3, 2)
    res[f"sb{sb}_extract_ms"] = round(ms / cnt, 3)
del os.environ["T9_EXTRACT_SB"]
print(json.dumps(res), flush=True)
nat.close()
