#!/usr/bin/env python3
"""Adversarial tie-throughput bench (VERDICT r01 item 2 'done' bar):
time t9_sort_records on 10 GiB of
  (a) uniform records (the normal bench input — reference time),
  (b) all-identical records (every byte equal: tie machinery runs but
      every tail chunk is identical -> zero pair sorts),
  (c) one shared 8-byte prefix, random tails (worst case: every tail
      chunk differs -> 12 chunk sorts + prefix pass at m = n).
Prints one JSON line with seconds per case and ratios vs uniform.
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
from tests import _gpu as G           # noqa: E402
from thrill_amd import Native         # noqa: E402

REC = 100
N = int(os.environ.get("T9_TIES_N", 10 * 1024**3 // REC))


CLASSES = ["extract", "hist_pairs", "pair_scatter", "lds_sort", "gather",
           "tie_partition", "tie_prescan"]


def time_sort(nat, din, dout, w, reps=3):
    s = G.stream()
    nat.sort_records(G.ptr(din), G.ptr(dout), N, REC, 10, G.ptr(w), s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        nat.sort_records(G.ptr(din), G.ptr(dout), N, REC, 10, G.ptr(w), s)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    # per-class breakdown from one extra instrumented run
    nat.perf_reset()
    nat.perf_enable(True)
    nat.sort_records(G.ptr(din), G.ptr(dout), N, REC, 10, G.ptr(w), s)
    torch.cuda.synchronize()
    nat.perf_enable(False)
    br = {}
    for c in CLASSES:
        ms, cnt = nat.perf_read(c)
        if cnt:
            br[c] = round(ms, 2)
    nat.perf_reset()
    return dt, br


def main():
    nat = Native(device=0)
    din = G.empty(N * REC, np.uint8)
    dout = G.empty(N * REC, np.uint8)
    w = G.ws(nat.ws("sort_records", N, REC))
    out = {"records": N}

    # (a) uniform
    nat.gen_records(G.ptr(din), 0, N, 0x7421, G.stream())
    out["uniform_s"], out["uniform_br"] = time_sort(nat, din, dout, w)
    out["uniform_s"] = round(out["uniform_s"], 4)

    # (b) all-identical records
    din.view(torch.uint8).fill_(0xA7)
    out["identical_s"], out["identical_br"] = time_sort(nat, din, dout, w)
    out["identical_s"] = round(out["identical_s"], 4)

    # (c) shared prefix, random tails: regenerate then stamp bytes 0..7
    nat.gen_records(G.ptr(din), 0, N, 0x7421, G.stream())
    v = din.view(N, REC)
    v[:, :8] = 0x55
    out["shared_prefix_s"], out["shared_prefix_br"] = \
        time_sort(nat, din, dout, w)
    out["shared_prefix_s"] = round(out["shared_prefix_s"], 4)

    out["identical_x"] = round(out["identical_s"] / out["uniform_s"], 2)
    out["shared_prefix_x"] = round(out["shared_prefix_s"] /
                                   out["uniform_s"], 2)
    # sanity: output of (c) must be sorted by full record on a sample
    got = dout.view(N, REC)[:: max(1, N // 2000)].cpu().numpy()
    rows = [tuple(r) for r in got]
    assert rows == sorted(rows), "adversarial output not sorted"
    print(json.dumps(out), flush=True)
    nat.close()


if __name__ == "__main__":
    main()
