#!/usr/bin/env python3
"""Minimal PMC target: run sort_pairs (variant from T9_PAIR_SCATTER) a few
times on 2^27 pairs so rocprofv3 --pmc can attribute counters to the
scatter kernel. No validation, no oracle."""
import os
import sys

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
for _ in range(int(os.environ.get("T9_PMC_REPS", "3"))):
    dk.copy_(src_k)
    dv.copy_(src_v)
    nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), s)
torch.cuda.synchronize()
print("done", flush=True)
nat.close()
