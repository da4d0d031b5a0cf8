#!/usr/bin/env python3
"""PMC target: 2 terasort steps at bench scale (for rocprofv3 --pmc
FETCH_SIZE / WRITE_SIZE attribution to the sort kernels)."""
import os, sys
import torch
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from thrill_amd.pipeline import TeraSort
ts = TeraSort(107_374_182, 0x7421, rank=0, world=1, device=0)
ts.generate()
torch.cuda.synchronize()
for _ in range(2):
    ts.step()
torch.cuda.synchronize()
print("done")
ts.close()
