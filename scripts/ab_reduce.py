#!/usr/bin/env python3
"""A/B helper for the reduce-build wave pre-combine: times the word_count
step at --n tokens over --vocab keys under the current T9_* env."""
import argparse
import os
import sys
import time

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from thrill_amd.pipeline import WordCount  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1 << 28)
    ap.add_argument("--vocab", type=int, default=1000)
    ap.add_argument("--zipf", type=float, default=1.1)
    ap.add_argument("--reps", type=int, default=4)
    args = ap.parse_args()
    wc = WordCount(args.n, args.vocab, args.zipf, seed=0x7, rank=0,
                   world=1, device=0, keys128=False)
    wc.generate()
    torch.cuda.synchronize()
    wc.step()
    torch.cuda.synchronize()
    times = []
    for _ in range(args.reps):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        wc.step()
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) * 1e3)
    ms = sorted(times)[len(times) // 2]
    print(f"n={args.n} vocab={args.vocab} zipf={args.zipf} "
          f"median_ms={ms:.2f} "
          f"({args.n / ms * 1e3 / 1e9:.1f} Gtok/s) "
          f"WAVECOMB={os.environ.get('T9_REDUCE_WAVECOMB','-')}")
    wc.close()


if __name__ == "__main__":
    main()
