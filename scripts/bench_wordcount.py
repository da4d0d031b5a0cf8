#!/usr/bin/env python3
"""Secondary measurement (not the driver bench line): ReduceByKey
word_count on 8 GiB of Zipf(1.1) tokens over a 10M vocabulary, single GPU
(BASELINE.json config 4 shape at N=1). Reports tokens/s and the
reduce_build kernel timing."""
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from thrill_amd.pipeline import WordCount  # noqa: E402

GIB = 1024 ** 3
n = 8 * GIB // 8          # 1,073,741,824 tokens (u64)
vocab = 10_000_000

wc = WordCount(n, vocab, 1.1, seed=0x44, rank=0, world=1, device=0)
print(f"# building zipf cdf ({vocab} entries) + generating "
      f"{n} tokens...", file=sys.stderr)
wc.generate()
torch.cuda.synchronize()

wc.step()  # warmup
torch.cuda.synchronize()
wc.nat.perf_reset()
wc.nat.perf_enable(True)
steps = 5
t0 = time.perf_counter()
for _ in range(steps):
    ok, ov, m = wc.step()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
wc.nat.perf_enable(False)
ms, cnt = wc.nat.perf_read("reduce_build")
# validation: counts must sum to n
total = int(ov.sum().item())
assert total == n, (total, n)
print(json.dumps({
    "metric": "ReduceByKey tokens/s (zipf1.1, 10M vocab)",
    "value": round(n * steps / dt, 1),
    "unit": "tokens/s",
    "n_gpus": 1,
    "steps": steps,
    "ms_per_step": round(dt / steps * 1e3, 2),
    "uniques": m,
    "reduce_build": {"avg_ms": round(ms / cnt, 3), "launches": cnt,
                     "algo_GBps_16B_per_token":
                     round(16.0 * n / (ms / cnt / 1e3) / 1e9, 1)},
    "data": "synthetic", "dtype": "u64",
}))
wc.close()
