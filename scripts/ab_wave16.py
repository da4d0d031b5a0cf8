#!/usr/bin/env python3
"""A/B timing helper for the wave16 tiers: times t9_sort_pairs_u64_u32 at
--n under whatever T9_* env is set, printing ms (median of --reps)."""
import argparse
import ctypes
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from thrill_amd import Native  # noqa: E402


def ptr(t):
    return ctypes.c_void_p(t.data_ptr())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1 << 27)
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()
    nat = Native(device=0)
    s = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    n = args.n
    dk0 = torch.empty(n, dtype=torch.int64, device="cuda")
    nat.gen_u64(ptr(dk0), 0, n, 7, s)
    dk = torch.empty_like(dk0)
    dv = torch.empty(n, dtype=torch.int32, device="cuda")
    w = torch.empty(nat.ws("sort_pairs", n), dtype=torch.uint8,
                    device="cuda")
    times = []
    for _ in range(args.reps + 2):
        dk.copy_(dk0)
        dv.copy_(torch.arange(n, dtype=torch.int32, device="cuda"))
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        nat.sort_pairs_u64_u32(ptr(dk), ptr(dv), n, ptr(w), s)
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) * 1e3)
    times = sorted(times[2:])
    signed = dk ^ (-2 ** 63)
    ok = bool((signed[1:] >= signed[:-1]).all().item())
    print(f"n={n} sorted={ok} median_ms={times[len(times)//2]:.2f} "
          f"env(W16MAX={os.environ.get('T9_WAVE16_MAX','-')},"
          f"SPAN16={os.environ.get('T9_SPAN_WAVE16','-')})")
    nat.close()


if __name__ == "__main__":
    main()
