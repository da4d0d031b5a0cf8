#!/usr/bin/env python3
"""bench.py — TeraSort benchmark per the driver contract.

Workload (BASELINE.json metric config): TeraSort of 10 GiB of 100-byte
records (107,374,182 records), synthetic seeded GenSort-style input
generated on-device. At N=1 the whole 10 GiB is sorted on one GPU (the
metric's configuration fits a single GPU); at N>1 the same total input is
sharded and each step runs the full distributed pipeline (sample ->
splitters -> classify -> partition -> RCCL all-to-all over xGMI -> local
sort). "strong" scaling: total work fixed as N grows.

A step = one complete sort of the input (input resident in HBM when the
timed region starts; the input buffer is read-only to the pipeline, so
every step redoes identical work). value = records sorted per second,
whole-job over all N ranks.

The roofline object tracks the dominant kernel (the payload gather —
out[i] = rec[idx[i]], ~45% of the step): achieved = algorithmic bytes per
launch (2R + 4 = 204 B per record) / HIP-event-
measured average launch duration, against the 8 TB/s HBM3E peak. traffic
is the PMC byte count per launch from the committed rocprofv3 calibration
under profiles/ (not measured live — PMC needs its own rocprofv3 pass).

cpu_baseline: the CPU oracle (reference-semantics restatement,
oracle/t9_oracle.cpp — kind "port") timed on the FULL workload on this
box's host cores (OpenMP; ~8 s for 10 GiB — within the bounded-sample
budget, so the whole thing is the sample; rank 0, N=1 only).
"""
import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

GIB = 1024 ** 3
REC = 100
N_RECORDS_10GIB = 10 * GIB // REC          # 107,374,182
SEED = 0x7421
PEAK_HBM_GBS = 8000.0                      # MI355X_MICROARCH.md spec peak


def log(msg):
    print(f"# {msg}", file=sys.stderr, flush=True)


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--records", type=int, default=N_RECORDS_10GIB)
    ap.add_argument("--seed", type=int, default=SEED)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    return ap.parse_args()


def validate(ts, d_out, n_out):
    """Cheap per-run integrity checks (full bit-parity is the test suite's
    job): key-sum conservation (permutation-invariant checksum) and
    non-decreasing u64 key prefixes of the output."""
    nat = ts.nat
    dk = torch.empty(max(n_out, 1), dtype=torch.int64, device="cuda")
    di = torch.empty(max(n_out, 1), dtype=torch.int32, device="cuda")
    if n_out:
        nat.extract_key64(ctypes.c_void_p(d_out.data_ptr()), n_out, REC, 0,
                          ctypes.c_void_p(dk.data_ptr()),
                          ctypes.c_void_p(di.data_ptr()),
                          ctypes.c_void_p(
                              torch.cuda.current_stream().cuda_stream))
    keysum = int(dk[:n_out].sum().item()) if n_out else 0
    signed = dk[:n_out] ^ (-2 ** 63)
    mono = bool((signed[1:] >= signed[:-1]).all().item()) if n_out > 1 \
        else True
    first = int(signed[0].item()) if n_out else None
    last = int(signed[-1].item()) if n_out else None
    return keysum, mono, first, last


def cpu_baseline_leg(seed, n_total):
    """Oracle (port of the reference CPU path: index sort under the
    full-record comparator + permute) on the FULL 10 GiB workload, OpenMP
    across the box's host cores."""
    from tests._oracle import Oracle
    o = Oracle()
    m = min(n_total, N_RECORDS_10GIB)
    recs = o.gen_records(m, seed=seed)
    t0 = time.perf_counter()
    _, cores = o.sort_records_parallel(recs)
    dt = time.perf_counter() - t0
    return {
        "value": m / dt,
        "unit": "keys/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{m} records ({m * REC / GIB:.2f} GiB) = the whole "
                  f"workload, oracle __gnu_parallel::sort index sort "
                  f"(full-record comparator) + permute, OpenMP over "
                  f"{cores} threads; {dt:.1f}s",
    }


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # control-plane backend: gloo (CPU scalars) by default, so that ALL
    # device traffic goes through the C-ABI t9_alltoall RCCL communicator
    # (the product data plane). T9_PG_BACKEND=nccl restores a torch RCCL
    # process group (needed for T9_EXCHANGE=torch).
    backend = os.environ.get("T9_PG_BACKEND", "gloo")
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        torch.cuda.set_device(local_rank)
        dist.init_process_group(backend)
        assert world == args.gpus, (world, args.gpus)
    elif os.environ.get("T9_FORCE_DIST"):
        # validation mode: run the distributed branch at world=1
        # (self-exchange) so the exact multi-rank code path is exercised
        import torch.distributed as dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29712")
        dist_mod.init_process_group(backend, rank=0, world_size=1)
    assert torch.cuda.is_available(), "bench.py needs a GPU (no CPU path)"

    from thrill_amd.pipeline import TeraSort
    n_total = args.records
    ts = TeraSort(n_total, args.seed, rank=rank, world=world,
                  device=local_rank)
    ts.generate()
    torch.cuda.synchronize()
    log(f"rank {rank}/{world}: generated {ts.n_local} records "
        f"({ts.n_local * REC / GIB:.2f} GiB) seed={hex(args.seed)}")

    # input key-sum for conservation check
    dk = torch.empty(ts.n_local, dtype=torch.int64, device="cuda")
    di = torch.empty(ts.n_local, dtype=torch.int32, device="cuda")
    ts.nat.extract_key64(ctypes.c_void_p(ts.d_in.data_ptr()), ts.n_local,
                         REC, 0, ctypes.c_void_p(dk.data_ptr()),
                         ctypes.c_void_p(di.data_ptr()),
                         ctypes.c_void_p(
                             torch.cuda.current_stream().cuda_stream))
    in_keysum = int(dk.sum().item())
    del dk, di

    for _ in range(args.warmup):
        ts.step()
    torch.cuda.synchronize()

    # roofline measurement pass (untimed, events on the product stream)
    ts.nat.perf_reset()
    ts.nat.perf_enable(True)
    out, n_out = ts.step()
    torch.cuda.synchronize()
    ts.nat.perf_enable(False)
    perf_breakdown = {}
    for cls in ["pair_scatter", "hist_pairs", "lds_sort", "extract", "gather"]:
        ms, cnt = ts.nat.perf_read(cls)
        perf_breakdown[cls] = {"total_ms": round(ms, 3), "launches": cnt}
    ts.nat.perf_reset()
    # The roofline tracks the DOMINANT kernel of the step: the payload
    # gather (out[i] = rec[idx[i]], k_gather_records_span — ~45% of the
    # step). Algorithmic bytes per launch: 100 B read
    # + 100 B written + 4 B index per record (SURVEY.md §8d "payload
    # gather = 2R"). Per-kernel shares for the rest are in
    # perf_breakdown.
    roofline = None
    g_ms, g_n = perf_breakdown["gather"]["total_ms"], \
        perf_breakdown["gather"]["launches"]
    if g_n:
        per_launch_s = g_ms / 1e3 / g_n
        algo_bytes = (2.0 * REC + 4.0) * ts.n_local
        achieved = algo_bytes / per_launch_s / 1e9
        # traffic: PMC bytes per launch from the committed calibration run
        # (profiles/pmc_traffic_terasort_r02.json, byte conversion
        # anchored on the gather's exact write size — separate rocprofv3
        # --pmc FETCH_SIZE / WRITE_SIZE passes; FETCH doubled per the
        # gfx950 half-reporting of wide coalesced reads, absolute values
        # carry the guide's per-pattern calibration caveat). Scaled to
        # this run's n. None when the calibration file is absent. The
        # gather's fetch traffic is ~2.4x the algorithmic read: random
        # 100-B record spans fetch whole cache lines (the measured
        # random-line wall — scripts/probe_gather_wall.py puts this
        # kernel at 82% of its own streaming rate).
        traffic = None
        cal_path = os.path.join(REPO, "profiles",
                                "pmc_traffic_terasort_r02.json")
        if not os.path.exists(cal_path):
            cal_path = os.path.join(REPO, "profiles",
                                    "pmc_traffic_terasort_r01.json")
        if os.path.exists(cal_path):
            import json as _json
            with open(cal_path) as f:
                cal = _json.load(f)
            tot = 0.0
            cnt = 0
            for k, v in cal.items():
                if "gather_records" in k:
                    tot += (v["fetch_bytes_x2_per_launch"] +
                            v["write_bytes_per_launch"])
                    cnt += 1
            if cnt:
                traffic = round(tot / cnt * (ts.n_local / 107_374_182))
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved, 1),
            "peak": PEAK_HBM_GBS,
            "unit": "GB/s",
            "frac": round(achieved / PEAK_HBM_GBS, 4),
            "traffic": traffic,
            "kernel": "gather (payload permute: out[i] = rec[idx[i]], "
                      "dominant kernel of the step)",
            "avg_launch_ms": round(g_ms / g_n, 3),
        }

    # timed region
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out, n_out = ts.step()
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if dist:
        ctl = "cpu" if dist.get_backend() == "gloo" else "cuda"
        t = torch.tensor([elapsed], device=ctl)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # validation
    keysum, mono, first, last = validate(ts, out, n_out)
    if dist:
        ctl = "cpu" if dist.get_backend() == "gloo" else "cuda"
        ksum = torch.tensor([keysum], dtype=torch.int64, device=ctl)
        dist.all_reduce(ksum)
        keysum = int(ksum.item())
        isum = torch.tensor([in_keysum], dtype=torch.int64, device=ctl)
        dist.all_reduce(isum)
        in_keysum = int(isum.item())
        # rank boundaries: my first key must be >= previous rank's last
        fl = [None] * world
        dist.all_gather_object(fl, (first, last))
        if rank == 0:
            for r in range(1, world):
                if fl[r][0] is not None and fl[r - 1][1] is not None:
                    assert fl[r][0] >= fl[r - 1][1], f"boundary {r}"
    assert mono, "output key prefixes not sorted"
    assert keysum == in_keysum, "key-sum conservation failed"

    ms_per_step = elapsed / args.steps * 1e3
    value = n_total * args.steps / elapsed

    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        log("timing CPU baseline (oracle, bounded sample)...")
        cpu_baseline = cpu_baseline_leg(args.seed, n_total)

    if rank == 0:
        result = {
            "metric": "TeraSort keys/sec (100-byte recs)",
            "value": round(value, 1),
            "unit": "keys/s",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": "terasort_10GiB_100B",
                "records": n_total,
                "record_bytes": REC,
                "seed": hex(args.seed),
                "parallelism": f"dp{args.gpus}",
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
            "perf_breakdown": perf_breakdown,
            "validated": {"keysum_conserved": True, "sorted": True},
        }
        print(json.dumps(result), flush=True)

    ts.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
