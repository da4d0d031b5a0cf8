#!/usr/bin/env python3
"""BASELINE config 5 at N=1: Sort<struct{u64 key; u8 payload[120]}> on
32 GiB = 268,435,456 x 128 B records (numeric key order, payload
tiebreak). Parity-config measurement, not the driver bench line."""
import ctypes, json, os, sys, time
import numpy as np, torch
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G
from thrill_amd import Native

nat = Native(device=0)
n = 268_435_456
REC = 128
s = G.stream()
din = G.empty(n * REC, np.uint8)
# fill with seeded random u64 words (keys = first 8 bytes of each record)
nat.gen_u64(G.ptr(din), 0, n * REC // 8, 0x5005, s)
dout = G.empty(n * REC, np.uint8)
w = G.ws(nat.ws("sort_records", n, REC))
torch.cuda.synchronize()
print(f"# generated {n} records (32 GiB)", file=sys.stderr)

nat.sort_records_keyle(G.ptr(din), G.ptr(dout), n, REC, G.ptr(w), s)
torch.cuda.synchronize()
ts = []
for _ in range(3):
    t0 = time.perf_counter()
    nat.sort_records_keyle(G.ptr(din), G.ptr(dout), n, REC, G.ptr(w), s)
    torch.cuda.synchronize()
    ts.append(time.perf_counter() - t0)
dt = min(ts)

# validate: keysum conservation + numeric sortedness of output keys
dk = torch.empty(n, dtype=torch.int64, device="cuda")
di = torch.empty(n, dtype=torch.int32, device="cuda")
nat.extract_key64_le(G.ptr(din), n, REC, 0, G.ptr(dk), G.ptr(di), s)
insum = int(dk.sum().item())
nat.extract_key64_le(G.ptr(dout), n, REC, 0, G.ptr(dk), G.ptr(di), s)
assert int(dk.sum().item()) == insum
signed = dk ^ (-2 ** 63)
assert bool((signed[1:] >= signed[:-1]).all().item())
print(json.dumps({
    "metric": "config5 Sort 128B-rec keys/s", "value": round(n / dt, 1),
    "unit": "keys/s", "n_gpus": 1, "ms_per_sort": round(dt * 1e3, 2),
    "records": n, "record_bytes": REC, "data": "synthetic",
    "validated": {"keysum_conserved": True, "sorted": True},
}))
nat.close()
