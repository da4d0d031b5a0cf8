#!/usr/bin/env python3
"""Measure radix scatter variants on the GPU (bench harness, not product).

Variants (env-selected in libt9):
  T9_PAIR_SCATTER: 1 = 2048-tile LDS-staged, 2 = 4096-tile global re-read,
                   3 = 8192-tile global re-read (1 WG/CU)
  T9_KEYS_SCATTER: 1 = 4096-tile staged, 2 = 8192-tile re-read
"""
import ctypes
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from tests import _gpu as G          # noqa: E402
from thrill_amd import Native        # noqa: E402


def time_sort(nat, fn, reps=3):
    ts = []
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    return min(ts)


def main():
    nat = Native(device=0)
    n = 1 << 27
    s = G.stream()

    # pristine input
    src_k = G.empty(n, np.uint64)
    nat.gen_u64(G.ptr(src_k), 0, n, 0x7421, s)
    src_v = torch.arange(n, dtype=torch.int32, device="cuda")
    dk = torch.empty_like(src_k)
    dv = torch.empty_like(src_v)
    w = G.ws(nat.ws("sort_pairs", n))
    torch.cuda.synchronize()

    results = {}
    for var in [4, 5, 6, 7]:
        os.environ["T9_PAIR_SCATTER"] = str(var)

        def run():
            dk.copy_(src_k)
            dv.copy_(src_v)
            nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), s)

        run()  # warmup + correctness
        torch.cuda.synchronize()
        signed = dk ^ (-2 ** 63)
        assert bool((signed[1:] >= signed[:-1]).all().item()), f"var{var}"
        assert int(dk.sum().item()) == int(src_k.sum().item())
        nat.perf_reset()
        nat.perf_enable(True)
        t = time_sort(nat, run)
        nat.perf_enable(False)
        ms, cnt = nat.perf_read("pair_scatter")
        nat.perf_reset()
        per_launch = ms / cnt
        algo = 24.0 * n
        results[f"pairs_v{var}"] = {
            "sort_s": round(t, 4),
            "scatter_avg_ms": round(per_launch, 3),
            "scatter_algo_GBps": round(algo / (per_launch / 1e3) / 1e9, 1),
            "Mpairs_per_s": round(n / t / 1e6, 1),
        }
        print(json.dumps({f"pairs_v{var}": results[f"pairs_v{var}"]}),
              flush=True)

    wk = G.ws(nat.ws("sort_u64", n))
    for var in [3, 4, 5]:
        os.environ["T9_KEYS_SCATTER"] = str(var)

        def runk():
            dk.copy_(src_k)
            nat.sort_u64(G.ptr(dk), n, G.ptr(wk), s)

        runk()
        torch.cuda.synchronize()
        signed = dk ^ (-2 ** 63)
        assert bool((signed[1:] >= signed[:-1]).all().item()), f"kvar{var}"
        nat.perf_reset()
        nat.perf_enable(True)
        t = time_sort(nat, runk)
        nat.perf_enable(False)
        ms, cnt = nat.perf_read("keys_scatter")
        nat.perf_reset()
        per_launch = ms / cnt
        algo = 16.0 * n
        results[f"keys_v{var}"] = {
            "sort_s": round(t, 4),
            "scatter_avg_ms": round(per_launch, 3),
            "scatter_algo_GBps": round(algo / (per_launch / 1e3) / 1e9, 1),
            "Mkeys_per_s": round(n / t / 1e6, 1),
        }
        print(json.dumps({f"keys_v{var}": results[f"keys_v{var}"]}),
              flush=True)

    nat.close()


if __name__ == "__main__":
    main()
