#!/usr/bin/env python3
"""Secondary measurement (not the driver bench line): ReduceByKey
word_count on 8 GiB of Zipf(1.1) tokens over a 10M vocabulary, single GPU
(BASELINE.json config 4 shape at N=1). Primary line = the 128-bit
composite-key path (string-identity semantics, the WordCount default);
the bare-u64 ReducePair path is reported beside it. Reports tokens/s and
the reduce_build kernel timing."""
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
n = 8 * GIB // 8          # 1,073,741,824 tokens (u64 ids)
vocab = 10_000_000


def measure(keys128, steps=5):
    wc = WordCount(n, vocab, 1.1, seed=0x44, rank=0, world=1, device=0,
                   keys128=keys128)
    print(f"# keys128={keys128}: generating {n} tokens...",
          file=sys.stderr)
    wc.generate()
    torch.cuda.synchronize()
    wc.step()  # warmup
    torch.cuda.synchronize()
    wc.nat.perf_reset()
    wc.nat.perf_enable(True)
    t0 = time.perf_counter()
    if keys128:
        for _ in range(steps):
            k1, k2, v, m = wc.step()
    else:
        for _ in range(steps):
            ok, v, m = wc.step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    wc.nat.perf_enable(False)
    ms, cnt = wc.nat.perf_read("reduce_build")
    total = int(v.sum().item())
    assert total == n, (total, n)
    out = {
        "keys128": keys128,
        "value": round(n * steps / dt, 1),
        "unit": "tokens/s",
        "ms_per_step": round(dt / steps * 1e3, 2),
        "uniques": m,
        "reduce_build": {"avg_ms": round(ms / cnt, 3), "launches": cnt},
    }
    wc.close()
    del wc
    torch.cuda.empty_cache()
    return out


res = {
    "metric": "ReduceByKey tokens/s (zipf1.1, 10M vocab)",
    "n_gpus": 1,
    "data": "synthetic",
    "pairs128": measure(True),
    "u64": measure(False),
}
print(json.dumps(res))
