#!/usr/bin/env python3
"""Root-cause probe for the round-1 anomaly: a 10.7 GB RCCL
self-exchange 'measured as a hang' (pipeline.py history, commit
551d307). Forces t9_alltoall through ncclSend/ncclRecv-to-self at
world=1 (T9_A2A_SELF=nccl overrides the loopback shortcut) at a given
size and times it. Run one size per process under `timeout` so a hang
is bounded and attributable:
    python scripts/probe_self_nccl.py <bytes>
The product path never self-sends (t9_alltoall moves the rank's own
share with a device copy), so this is evidence-gathering only.
"""
import os
import sys
import time

os.environ["T9_A2A_SELF"] = "nccl"   # before the first t9_alltoall call

import numpy as np                    # noqa: E402
import torch                          # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G           # noqa: E402
from thrill_amd import Native         # noqa: E402
import ctypes                         # noqa: E402

nbytes = int(sys.argv[1]) if len(sys.argv) > 1 else 1 << 30
nat = Native(device=0, rank=0, world=1)
nat.comm_init(nat.comm_id())
ds = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
dr = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
ds.fill_(7)
c = np.array([nbytes], dtype=np.uint64)
d = np.array([0], dtype=np.uint64)
t0 = time.perf_counter()
nat.alltoall(G.ptr(ds), ctypes.c_void_p(c.ctypes.data),
             ctypes.c_void_p(d.ctypes.data), G.ptr(dr),
             ctypes.c_void_p(c.ctypes.data),
             ctypes.c_void_p(d.ctypes.data), 1, G.stream())
torch.cuda.synchronize()
dt = time.perf_counter() - t0
ok = bool((dr == 7).all().item())
print(f'{{"bytes": {nbytes}, "seconds": {dt:.3f}, "ok": {str(ok).lower()},'
      f' "GBps": {nbytes / dt / 1e9:.1f}}}', flush=True)
nat.close()
