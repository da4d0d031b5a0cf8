"""GPU parity tests: the HIP path (libt9 C ABI) vs the CPU oracle on
identical seeded inputs. Bit-exact everywhere (integer/byte work).

Parity definition: Sort — byte-identical output under the
acceptance total order; classification — element-wise equal bucket ids.
"""
import ctypes

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from tests import _gpu as G
    from thrill_amd import Native


@pytest.fixture(scope="module")
def nat():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    n = Native(device=0)
    yield n
    n.close()


# ------------------------------------------------------------- generators

def test_gen_u64_matches_oracle(nat, oracle):
    n = 1 << 20
    d = G.empty(n, np.uint64)
    nat.gen_u64(G.ptr(d), 0, n, 0x7421, G.stream())
    got = G.host(d, np.uint64)
    assert np.array_equal(got, oracle.gen_u64(n, seed=0x7421))


def test_gen_u64_offset_matches_oracle(nat, oracle):
    n = 4096
    d = G.empty(n, np.uint64)
    nat.gen_u64(G.ptr(d), 1000, n, 5, G.stream())
    assert np.array_equal(G.host(d, np.uint64),
                          oracle.gen_u64(n, seed=5, index0=1000))


def test_gen_records_matches_oracle(nat, oracle):
    n = 100_000
    d = G.empty(n * 100, np.uint8)
    nat.gen_records(G.ptr(d), 0, n, 42, G.stream())
    got = G.host(d, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.gen_records(n, seed=42))


# ------------------------------------------------------------- u64 sort

@pytest.mark.parametrize("n", [0, 1, 2, 63, 64, 255, 4096, 4097,
                               1_000_003, 1 << 22])
def test_sort_u64_parity(nat, oracle, n):
    keys = oracle.gen_u64(n, seed=n + 1) if n else np.empty(0, np.uint64)
    d = G.dev(keys) if n else G.empty(0, np.uint64)
    w = G.ws(nat.ws("sort_u64", n))
    nat.sort_u64(G.ptr(d), n, G.ptr(w), G.stream())
    got = G.host(d, np.uint64)
    assert np.array_equal(got, np.sort(keys))


def test_sort_u64_known_integer_identity(nat):
    # sort_node_test.cpp:25-53 restated: reversed known integers -> identity
    n = 1 << 21
    keys = np.arange(n - 1, -1, -1, dtype=np.uint64)
    d = G.dev(keys)
    w = G.ws(nat.ws("sort_u64", n))
    nat.sort_u64(G.ptr(d), n, G.ptr(w), G.stream())
    assert np.array_equal(G.host(d, np.uint64),
                          np.arange(n, dtype=np.uint64))


@pytest.mark.parametrize("case", ["all_equal", "few_values", "presorted",
                                  "reverse_u32_range", "high_bits_only"])
def test_sort_u64_distributions(nat, case):
    n = 1 << 18
    rng = np.random.default_rng(7)
    if case == "all_equal":
        keys = np.full(n, 0xDEADBEEF, dtype=np.uint64)
    elif case == "few_values":
        keys = rng.integers(0, 4, n).astype(np.uint64)
    elif case == "presorted":
        keys = np.sort(rng.integers(0, 1 << 63, n).astype(np.uint64))
    elif case == "reverse_u32_range":
        keys = np.arange(n, dtype=np.uint64)[::-1].copy()
    else:
        keys = (rng.integers(0, 1 << 16, n).astype(np.uint64)) << 48
    d = G.dev(keys)
    w = G.ws(nat.ws("sort_u64", n))
    nat.sort_u64(G.ptr(d), n, G.ptr(w), G.stream())
    assert np.array_equal(G.host(d, np.uint64), np.sort(keys))


# ------------------------------------------------------------- pair sort

def test_sort_pairs_stability(nat):
    # stable by construction: equal keys keep ascending payload order
    n = 1 << 18
    rng = np.random.default_rng(3)
    keys = rng.integers(0, 64, n).astype(np.uint64)
    vals = np.arange(n, dtype=np.uint32)
    dk, dv = G.dev(keys), G.dev(vals)
    w = G.ws(nat.ws("sort_pairs", n))
    nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), G.stream())
    gk, gv = G.host(dk, np.uint64), G.host(dv, np.uint32)
    assert np.array_equal(gk, np.sort(keys))
    # numpy stable argsort is the stability oracle
    order = np.argsort(keys, kind="stable").astype(np.uint32)
    assert np.array_equal(gv, order)


# ------------------------------------------------------------- record sort

@pytest.mark.parametrize("n", [0, 1, 2, 1000, 100_000])
def test_sort_records_parity(nat, oracle, n):
    recs = oracle.gen_records(n, seed=n + 9) if n \
        else np.empty((0, 100), np.uint8)
    din = G.dev(recs.reshape(-1)) if n else G.empty(0, np.uint8)
    dout = G.empty(max(n, 1) * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    if n:
        got = G.host(dout, np.uint8)[:n * 100].reshape(n, 100)
        assert np.array_equal(got, oracle.sort_records(recs))


def test_sort_records_equal_key_ties(nat, oracle):
    # adversarial: many records share the full 10-byte key; acceptance
    # order = full-record lexicographic (ties resolved by value bytes).
    n = 4096
    recs = oracle.gen_records(n, seed=1)
    recs[:, :10] = 0x55            # all keys identical
    rng = np.random.default_rng(2)
    recs[:, 10:] = rng.integers(0, 256, (n, 90)).astype(np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


def test_sort_records_prefix_collisions(nat, oracle):
    # records with equal u64 prefixes but different key bytes 8..9:
    # exercises the tie-fix path's key-tail ordering.
    n = 2048
    recs = oracle.gen_records(n, seed=4)
    recs[:, :8] = np.tile(np.arange(16, dtype=np.uint8), (n, 1))[:, :8]
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


def test_sort_records_fused_extract_hist(nat, oracle):
    # the fused extract+pass1-hist path (default at MSB-dispatch sizes);
    # engage it at test scale via T9_SORT_ALGO=msb
    import os
    os.environ["T9_SORT_ALGO"] = "msb"
    try:
        n = 300_000
        recs = oracle.gen_records(n, seed=77)
        din = G.dev(recs.reshape(-1))
        dout = G.empty(n * 100, np.uint8)
        w = G.ws(nat.ws("sort_records", n, 100))
        nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                         G.stream())
        got = G.host(dout, np.uint8).reshape(n, 100)
        assert np.array_equal(got, oracle.sort_records(recs))
        # and with the fusion disabled: identical result
        os.environ["T9_EXTRACT_HIST"] = "0"
        nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                         G.stream())
        got2 = G.host(dout, np.uint8).reshape(n, 100)
        assert np.array_equal(got2, got)
    finally:
        del os.environ["T9_SORT_ALGO"]
        os.environ.pop("T9_EXTRACT_HIST", None)


def test_sort_records_keyle_fused_extract(nat):
    # the fused LE extract+hist path (config 5's default at scale),
    # engaged at test size via T9_SORT_ALGO=msb
    import os
    os.environ["T9_SORT_ALGO"] = "msb"
    try:
        n = 200_000
        rng = np.random.default_rng(91)
        recs = rng.integers(0, 256, (n, 128)).astype(np.uint8)
        recs[: n // 20, :8] = recs[0, :8]   # duplicate numeric keys
        din = G.dev(recs.reshape(-1))
        dout = G.empty(n * 128, np.uint8)
        w = G.ws(nat.ws("sort_records", n, 128))
        nat.sort_records_keyle(G.ptr(din), G.ptr(dout), n, 128, G.ptr(w),
                               G.stream())
        got = G.host(dout, np.uint8).reshape(n, 128)
        keys = recs[:, :8].copy().view("<u8").reshape(-1)
        order = np.lexsort(tuple(recs[:, c] for c in range(127, 7, -1))
                           + (keys,))
        assert np.array_equal(got, recs[order])
    finally:
        del os.environ["T9_SORT_ALGO"]


def test_gather_scatter_variant_parity(nat, oracle):
    # T9_GATHER_VARIANT=5 (sequential-read random-write scatter probe)
    import os
    n = 200_000
    recs = oracle.gen_records(n, seed=88)
    rng = np.random.default_rng(88)
    idx = rng.permutation(n).astype(np.uint32)
    din, didx = G.dev(recs.reshape(-1)), G.dev(idx)
    dout = G.empty(n * 100, np.uint8)
    os.environ["T9_GATHER_VARIANT"] = "5"
    try:
        nat.gather_records(G.ptr(din), G.ptr(didx), n, 100, G.ptr(dout),
                           G.stream())
    finally:
        del os.environ["T9_GATHER_VARIANT"]
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, recs[idx])


def test_sort_records_many_tie_runs(nat, oracle):
    # the on-device tie sort's regrouping pass: many distinct equal-prefix
    # runs interleaved with unique-prefix records; the final stable
    # prefix pass must restore run grouping after the tail-chunk LSD.
    n = 200_000
    rng = np.random.default_rng(7)
    recs = oracle.gen_records(n, seed=11)
    # 1000 shared prefixes over half the records, the rest unique
    share = rng.choice(n, n // 2, replace=False)
    pref = rng.integers(0, 1000, len(share))
    for b in range(8):
        recs[share, b] = ((pref >> (8 * (7 - b))) & 0xFF).astype(np.uint8)
    recs[share, 10:] = rng.integers(0, 256, (len(share), 90)) \
        .astype(np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


def test_sort_records_msd_fallback_low_entropy_tails(nat, oracle):
    # tie pattern that PERSISTS past 2 MSD levels: one shared prefix,
    # tail chunks drawn from {0,1} — every chunk differs globally but
    # most pairs still tie after each level, so the bounded-LSD fallback
    # runs. Also cross-checks T9_TIE_MSD=0 (pure LSD) bit-for-bit.
    import os
    n = 50_000
    rng = np.random.default_rng(17)
    recs = np.zeros((n, 100), np.uint8)
    recs[:, :10] = 0x42
    for b in (15, 23, 31, 47, 80):
        recs[:, b] = rng.integers(0, 2, n).astype(np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    expect = oracle.sort_records(recs)
    assert np.array_equal(got, expect)
    os.environ["T9_TIE_MSD"] = "0"
    try:
        nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                         G.stream())
        got2 = G.host(dout, np.uint8).reshape(n, 100)
        assert np.array_equal(got2, expect)
    finally:
        del os.environ["T9_TIE_MSD"]


@pytest.mark.parametrize("seed", [101, 202, 303, 404])
def test_sort_records_tie_fuzz(nat, oracle, seed):
    # randomized duplicate structure: random prefix pool sizes, random
    # low-entropy tail bytes — sweeps MSD levels 0/1/2 and the LSD
    # fallback across seeds; oracle-exact.
    rng = np.random.default_rng(seed)
    n = int(rng.integers(5_000, 60_000))
    recs = oracle.gen_records(n, seed=seed)
    pool = int(rng.integers(1, 50))
    prefixes = rng.integers(0, pool, n)
    for b in range(8):
        recs[:, b] = ((prefixes >> max(0, 8 * (3 - b))) & 0xFF) \
            .astype(np.uint8)
    # tails: a random subset of byte columns collapsed to tiny alphabets
    for b in rng.choice(np.arange(10, 100), rng.integers(3, 20),
                        replace=False):
        recs[:, b] = rng.integers(0, int(rng.integers(2, 5)), n) \
            .astype(np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


def test_sort_records_all_identical(nat, oracle):
    # every byte of every record equal: the tie sort's chunk-skip path
    # (no chunk differs, so zero pair sorts run; stable order preserved)
    n = 50_000
    recs = np.full((n, 100), 0xA7, np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, recs)


def test_sort_records_narrow_ties(nat, oracle):
    # rec_size=12 (< 24 B): the tie scratch cannot fit in d_out, so the
    # temporary-allocation fallback runs. Ties differ only in the last
    # 4 tail bytes (a partial 8-byte chunk, zero-padded).
    n = 30_000
    rng = np.random.default_rng(13)
    recs = rng.integers(0, 256, (n, 12)).astype(np.uint8)
    recs[:, :8] = np.array([1, 2, 3, 4, 5, 6, 7, 8], np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 12, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 12))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 12, 12, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 12)
    order = np.lexsort(tuple(recs[:, c] for c in range(11, -1, -1)))
    assert np.array_equal(got, recs[order])


def test_extract_key64_bigendian(nat, oracle):
    n = 10_000
    recs = oracle.gen_records(n, seed=6)
    din = G.dev(recs.reshape(-1))
    dk, di = G.empty(n, np.uint64), G.empty(n, np.uint32)
    nat.extract_key64(G.ptr(din), n, 100, 0, G.ptr(dk), G.ptr(di),
                      G.stream())
    gk = G.host(dk, np.uint64)
    expect = np.array([int.from_bytes(r[:8].tobytes(), "big")
                       for r in recs], dtype=np.uint64)
    assert np.array_equal(gk, expect)
    assert np.array_equal(G.host(di, np.uint32),
                          np.arange(n, dtype=np.uint32))


# ------------------------------------------------------------- classify

@pytest.mark.parametrize("p", [2, 3, 5, 8])
def test_classify_parity(nat, oracle, p):
    n = 200_000
    rng = np.random.default_rng(p)
    # duplicate-heavy keys to exercise the (key, gidx) tiebreak
    keys = rng.integers(0, 97, n).astype(np.uint64)
    pos = rng.choice(n, 128, replace=False).astype(np.uint64)
    sk = keys[pos.astype(np.int64)]
    spl_k, spl_i = oracle.select_splitters_u64(sk, pos, p)
    gidx0 = 10_000
    expect = oracle.classify_u64(keys, gidx0, spl_k, spl_i, p)

    dk = G.dev(keys)
    dsk, dsi = G.dev(spl_k), G.dev(spl_i)
    db = G.empty(n, np.uint32)
    dc = G.empty(p, np.uint64)
    nat.classify_u64(G.ptr(dk), n, gidx0, G.ptr(dsk), G.ptr(dsi), p,
                     G.ptr(db), G.ptr(dc), G.stream())
    got = G.host(db, np.uint32)
    assert np.array_equal(got, expect)
    counts = G.host(dc, np.uint64)
    assert np.array_equal(counts, np.bincount(expect, minlength=p)
                          .astype(np.uint64))


@pytest.mark.parametrize("p", [2, 8])
def test_partition_idx_stable(nat, oracle, p):
    n = 100_000
    rng = np.random.default_rng(p + 50)
    bucket = rng.integers(0, p, n).astype(np.uint32)
    db = G.dev(bucket)
    dperm = G.empty(n, np.uint32)
    doffs = G.empty(p + 1, np.uint64)
    w = G.ws(nat.ws("partition_idx", n))
    nat.partition_idx(G.ptr(db), n, p, G.ptr(dperm), G.ptr(doffs),
                      G.ptr(w), G.stream())
    perm = G.host(dperm, np.uint32)
    offs = G.host(doffs, np.uint64)
    expect_offs = np.concatenate(
        [[0], np.cumsum(np.bincount(bucket, minlength=p))]).astype(np.uint64)
    assert np.array_equal(offs, expect_offs)
    # stable grouping: within each bucket original order is preserved
    expect_perm = np.argsort(bucket, kind="stable").astype(np.uint32)
    assert np.array_equal(perm, expect_perm)


def test_end_to_end_single_gpu_sample_sort(nat, oracle):
    """Full single-GPU pipeline at p=4 virtual buckets: classify ->
    partition -> per-bucket sort -> concat == total sort (the reference
    MainOp semantics, api/sort.hpp:537-663)."""
    n = 500_000
    keys = oracle.gen_u64(n, seed=99)
    # worker-style samples: every k-th key, global indices = positions
    pos = np.arange(0, n, n // 512, dtype=np.uint64)[:512]
    p = 4
    spl_k, spl_i = oracle.select_splitters_u64(
        keys[pos.astype(np.int64)], pos, p)
    dk = G.dev(keys)
    dsk, dsi = G.dev(spl_k), G.dev(spl_i)  # keep alive: kernels are queued
    db = G.empty(n, np.uint32)
    dc = G.empty(p, np.uint64)
    nat.classify_u64(G.ptr(dk), n, 0, G.ptr(dsk), G.ptr(dsi), p,
                     G.ptr(db), G.ptr(dc), G.stream())
    dperm = G.empty(n, np.uint32)
    doffs = G.empty(p + 1, np.uint64)
    w = G.ws(max(nat.ws("partition_idx", n), nat.ws("sort_u64", n)))
    nat.partition_idx(G.ptr(db), n, p, G.ptr(dperm), G.ptr(doffs),
                      G.ptr(w), G.stream())
    perm = G.host(dperm, np.uint32)
    offs = G.host(doffs, np.uint64)
    grouped = keys[perm.astype(np.int64)]
    out = []
    for b in range(p):
        part = grouped[int(offs[b]):int(offs[b + 1])]
        dpart = G.dev(part)
        nat.sort_u64(G.ptr(dpart), len(part), G.ptr(w), G.stream())
        out.append(G.host(dpart, np.uint64))
    assert np.array_equal(np.concatenate(out), np.sort(keys))


def test_alltoall_loopback_world1(nat):
    # t9_alltoall's world==1 self-exchange path (the RCCL grouped path is
    # exercised by the driver's multi-GPU run; counts/displs marshalling
    # is shared)
    import ctypes
    n = 10_000
    send = G.dev(np.arange(n, dtype=np.uint64))
    recv = G.empty(n, np.uint64)
    counts = np.array([n], dtype=np.uint64)
    displs = np.array([0], dtype=np.uint64)
    cp = counts.ctypes.data_as(ctypes.c_void_p)
    dp = displs.ctypes.data_as(ctypes.c_void_p)
    nat.alltoall(G.ptr(send), cp, dp, G.ptr(recv), cp, dp, 8, G.stream())
    assert np.array_equal(G.host(recv, np.uint64),
                          np.arange(n, dtype=np.uint64))


@pytest.mark.parametrize("na,nb", [(0, 100), (100, 0), (1000, 1),
                                   (100_000, 250_001), (1 << 20, 1 << 20)])
def test_merge_u64_parity(nat, oracle, na, nb):
    a = np.sort(oracle.gen_u64(na, seed=na + 1)) if na \
        else np.empty(0, np.uint64)
    b = np.sort(oracle.gen_u64(nb, seed=nb + 2)) if nb \
        else np.empty(0, np.uint64)
    da = G.dev(a) if na else G.empty(0, np.uint64)
    db = G.dev(b) if nb else G.empty(0, np.uint64)
    dout = G.empty(na + nb, np.uint64)
    nat.merge_u64(G.ptr(da), na, G.ptr(db), nb, G.ptr(dout), G.stream())
    got = G.host(dout, np.uint64)
    assert np.array_equal(got, np.sort(np.concatenate([a, b]),
                                       kind="stable"))


def test_merge_u64_tie_source_order(nat):
    # equal keys: all of a's precede b's — distinguish via known layout
    a = np.array([5, 5, 7], dtype=np.uint64)
    b = np.array([5, 6, 7, 7], dtype=np.uint64)
    da, db = G.dev(a), G.dev(b)
    dout = G.empty(7, np.uint64)
    nat.merge_u64(G.ptr(da), 3, G.ptr(db), 4, G.ptr(dout), G.stream())
    assert G.host(dout, np.uint64).tolist() == [5, 5, 5, 6, 7, 7, 7]


@pytest.mark.parametrize("seed", [11, 222, 3333, 44444, 0xDEAD])
def test_sort_records_parity_seed_sweep(nat, oracle, seed):
    n = 50_000
    recs = oracle.gen_records(n, seed=seed)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


@pytest.mark.parametrize("rec", [100, 128, 12])
def test_merge_records_parity(nat, oracle, rec):
    # t9_merge_records: byte-lex merge of two sorted record sequences,
    # A-wins ties (reference api/merge.hpp source-order rule)
    rng = np.random.default_rng(rec)
    na, nb = 40_000, 55_000
    A = rng.integers(0, 4, (na, rec)).astype(np.uint8)  # tiny alphabet:
    B = rng.integers(0, 4, (nb, rec)).astype(np.uint8)  # many equals
    A = A[np.lexsort(tuple(A[:, c] for c in range(rec - 1, -1, -1)))]
    B = B[np.lexsort(tuple(B[:, c] for c in range(rec - 1, -1, -1)))]
    da, db = G.dev(A.reshape(-1)), G.dev(B.reshape(-1))
    dout = G.empty((na + nb) * rec, np.uint8)
    nat.merge_records(G.ptr(da), na, G.ptr(db), nb, rec, G.ptr(dout),
                      G.stream())
    got = G.host(dout, np.uint8).reshape(na + nb, rec)
    # expected: stable merge with A-before-B on equals == merging
    # (record, source) pairs sorted by (record, source)
    tagged = np.concatenate([A, B])
    src = np.concatenate([np.zeros(na, np.uint8), np.ones(nb, np.uint8)])
    order = np.lexsort((src,) + tuple(tagged[:, c]
                                      for c in range(rec - 1, -1, -1)))
    # within equal (record, source), keep original order: lexsort stable
    assert np.array_equal(got, tagged[order])


def test_merge_records_empty_sides(nat, oracle):
    rec = 100
    recs = oracle.gen_records(1000, seed=3)
    s = recs[np.lexsort(tuple(recs[:, c] for c in range(99, -1, -1)))]
    da = G.dev(s.reshape(-1))
    dout = G.empty(1000 * rec, np.uint8)
    nat.merge_records(G.ptr(da), 1000, G.ptr(da), 0, rec, G.ptr(dout),
                      G.stream())
    assert np.array_equal(G.host(dout, np.uint8).reshape(1000, rec), s)
    nat.merge_records(G.ptr(da), 0, G.ptr(da), 1000, rec, G.ptr(dout),
                      G.stream())
    assert np.array_equal(G.host(dout, np.uint8).reshape(1000, rec), s)
