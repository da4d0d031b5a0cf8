"""GPU parity tests for the two-level MSB radix path (T9_SORT_ALGO=msb),
including the skew fallbacks (oversize sub-buckets -> ranged LSD; equal
low-48-bit sub-buckets -> skip)."""
import os

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


@pytest.fixture(autouse=True)
def msb_env():
    os.environ["T9_SORT_ALGO"] = "msb"
    os.environ["T9_FUSED_EXTRACT"] = "1"   # record tests cover both paths
    yield
    del os.environ["T9_SORT_ALGO"]
    del os.environ["T9_FUSED_EXTRACT"]


def sort_pairs(nat, keys, vals):
    n = len(keys)
    dk, dv = G.dev(keys), G.dev(vals)
    w = G.ws(nat.ws("sort_pairs", n))
    nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w), G.stream())
    return G.host(dk, np.uint64), G.host(dv, np.uint32)


def check_stable(keys, vals, gk, gv):
    assert np.array_equal(gk, np.sort(keys))
    order = np.argsort(keys, kind="stable").astype(np.uint32)
    assert np.array_equal(gv, vals[order.astype(np.int64)])


@pytest.mark.parametrize("n", [1 << 14, 100_001, 1 << 20, (1 << 22) + 17])
def test_msb_uniform_parity_and_stability(nat, oracle, n):
    keys = oracle.gen_u64(n, seed=n)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_all_equal_keys(nat):
    # one (b7,b6) sub-bucket holds everything -> oversize -> ranged LSD;
    # and the low-48 equal-skip path inside it
    n = 1 << 18
    keys = np.full(n, 0xABCDEF0123456789, dtype=np.uint64)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_equal_top16_varying_low(nat):
    # single oversize sub-bucket with differing low bits -> ranged LSD
    n = 1 << 18
    rng = np.random.default_rng(1)
    keys = (np.uint64(0x7777) << np.uint64(48)) | \
        rng.integers(0, 1 << 48, n).astype(np.uint64)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_equal_low48(nat):
    # sub-buckets all take the equal-low-48 skip path
    n = 1 << 18
    rng = np.random.default_rng(2)
    keys = rng.integers(0, 1 << 16, n).astype(np.uint64) << np.uint64(48)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_few_values(nat):
    # 8 distinct keys: 8 sub-buckets oversize (ranged LSD each, <=64)
    n = 1 << 18
    rng = np.random.default_rng(3)
    keys = rng.integers(0, 8, n).astype(np.uint64) * np.uint64(2**61)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_moderate_skew_many_oversize(nat):
    # 1024 distinct top-16 values at n=2^20 -> ~1024 oversize sub-buckets
    # (> 64) -> full-LSD fallback path
    n = 1 << 20
    rng = np.random.default_rng(4)
    keys = (rng.integers(0, 1024, n).astype(np.uint64) << np.uint64(48)) | \
        rng.integers(0, 1 << 30, n).astype(np.uint64)
    vals = np.arange(n, dtype=np.uint32)
    gk, gv = sort_pairs(nat, keys, vals)
    check_stable(keys, vals, gk, gv)


def test_msb_records_end_to_end(nat, oracle):
    # record sort routed through MSB (sort_records -> sort_pairs dispatch)
    n = 300_000
    recs = oracle.gen_records(n, seed=77)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 100, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 100))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 100)
    assert np.array_equal(got, oracle.sort_records(recs))


@pytest.mark.parametrize("case", ["uniform", "all_equal", "few_values",
                                  "high_bits_only", "low48_equal"])
def test_msb_keys_only(nat, oracle, case):
    n = 1 << 18
    rng = np.random.default_rng(hash(case) % 2**31)
    if case == "uniform":
        keys = oracle.gen_u64(n, seed=1)
    elif case == "all_equal":
        keys = np.full(n, 0x123456789ABCDEF0, dtype=np.uint64)
    elif case == "few_values":
        keys = rng.integers(0, 5, n).astype(np.uint64) * np.uint64(2**62)
    elif case == "high_bits_only":
        keys = rng.integers(0, 1 << 16, n).astype(np.uint64) << np.uint64(48)
    else:
        keys = rng.integers(0, 1 << 30, n).astype(np.uint64)
    d = G.dev(keys)
    w = G.ws(nat.ws("sort_u64", n))
    nat.sort_u64(G.ptr(d), n, G.ptr(w), G.stream())
    assert np.array_equal(G.host(d, np.uint64), np.sort(keys))


def test_msb_records_128B_fused(nat):
    # config-5-shaped records (128 B) through the fused-extract MSB path;
    # acceptance order = full-record byte order (big-endian u64 prefix)
    n = 100_000
    rng = np.random.default_rng(31)
    recs = rng.integers(0, 256, (n, 128)).astype(np.uint8)
    din = G.dev(recs.reshape(-1))
    dout = G.empty(n * 128, np.uint8)
    w = G.ws(nat.ws("sort_records", n, 128))
    nat.sort_records(G.ptr(din), G.ptr(dout), n, 128, 8, G.ptr(w),
                     G.stream())
    got = G.host(dout, np.uint8).reshape(n, 128)
    order = np.lexsort(tuple(recs[:, c] for c in range(127, -1, -1)))
    assert np.array_equal(got, recs[order])


@pytest.mark.parametrize("n", [(1 << 22) - 1, (1 << 22), (1 << 22) + 1])
def test_dispatch_boundary_sizes(nat, n):
    # the LSD/MSB dispatch threshold (2^22) must be seamless
    del os.environ["T9_SORT_ALGO"]   # default dispatch
    try:
        dk = G.empty(n, np.uint64)
        nat.gen_u64(G.ptr(dk), 0, n, n, G.stream())
        insum = int(dk.sum().item())
        w = G.ws(nat.ws("sort_u64", n))
        nat.sort_u64(G.ptr(dk), n, G.ptr(w), G.stream())
        assert int(dk.sum().item()) == insum
        signed = dk ^ (-2 ** 63)
        assert bool((signed[1:] >= signed[:-1]).all().item())
    finally:
        os.environ["T9_SORT_ALGO"] = "msb"


@pytest.mark.parametrize("case", ["uniform", "few_values", "all_equal",
                                  "low_entropy_top"])
def test_three_level_forced_parity(nat, oracle, case):
    """3-level MSB (T9_MSB_LEVELS=3) at test size, vs stable argsort."""
    os.environ["T9_MSB_LEVELS"] = "3"
    try:
        n = 1 << 18
        rng = np.random.default_rng(len(case))
        if case == "uniform":
            keys = oracle.gen_u64(n, seed=2)
        elif case == "few_values":
            keys = rng.integers(0, 6, n).astype(np.uint64) * np.uint64(2**60)
        elif case == "all_equal":
            keys = np.full(n, 0x0102030405060708, dtype=np.uint64)
        else:  # varies only below the top 24 bits
            keys = rng.integers(0, 1 << 40, n).astype(np.uint64)
        vals = np.arange(n, dtype=np.uint32)
        dk, dv = G.dev(keys), G.dev(vals)
        w = G.ws(nat.ws("sort_pairs", n))
        nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w),
                               G.stream())
        gk, gv = G.host(dk, np.uint64), G.host(dv, np.uint32)
        assert np.array_equal(gk, np.sort(keys))
        order = np.argsort(keys, kind="stable").astype(np.uint32)
        assert np.array_equal(gv, vals[order.astype(np.int64)])
    finally:
        del os.environ["T9_MSB_LEVELS"]


def test_three_level_forced_records(nat, oracle):
    os.environ["T9_MSB_LEVELS"] = "3"
    try:
        n = 200_000
        recs = oracle.gen_records(n, seed=55)
        din = G.dev(recs.reshape(-1))
        dout = G.empty(n * 100, np.uint8)
        w = G.ws(nat.ws("sort_records", n, 100))
        nat.sort_records(G.ptr(din), G.ptr(dout), n, 100, 10, G.ptr(w),
                         G.stream())
        got = G.host(dout, np.uint8).reshape(n, 100)
        assert np.array_equal(got, oracle.sort_records(recs))
    finally:
        del os.environ["T9_MSB_LEVELS"]


def test_sort_records_keyle_config5_shape(nat):
    """config 5 semantics: 128 B records, native u64 key, numeric order,
    payload-byte tiebreak. The module's msb_env fixture forces the MSB
    dispatch, so this also exercises the fused LE extract+hist path (the
    default at config-5 sizes)."""
    n = 150_000
    rng = np.random.default_rng(41)
    recs = rng.integers(0, 256, (n, 128)).astype(np.uint8)
    # plant duplicate numeric keys to exercise the payload tiebreak
    recs[: n // 10, :8] = recs[0, :8]
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


@pytest.mark.parametrize("case", ["uniform", "few_values", "skew_top"])
def test_pass2_9bit_parity(nat, oracle, case):
    """experimental 9-bit level-2 digit (T9_PASS2_BITS=9)."""
    os.environ["T9_PASS2_BITS"] = "9"
    try:
        n = 1 << 18
        rng = np.random.default_rng(len(case) + 7)
        if case == "uniform":
            keys = oracle.gen_u64(n, seed=77)
        elif case == "few_values":
            keys = rng.integers(0, 9, n).astype(np.uint64) * np.uint64(2**59)
        else:
            keys = rng.integers(0, 1 << 30, n).astype(np.uint64)
        vals = np.arange(n, dtype=np.uint32)
        dk, dv = G.dev(keys), G.dev(vals)
        w = G.ws(nat.ws("sort_pairs", n))
        nat.sort_pairs_u64_u32(G.ptr(dk), G.ptr(dv), n, G.ptr(w),
                               G.stream())
        gk, gv = G.host(dk, np.uint64), G.host(dv, np.uint32)
        assert np.array_equal(gk, np.sort(keys))
        order = np.argsort(keys, kind="stable").astype(np.uint32)
        assert np.array_equal(gv, vals[order.astype(np.int64)])
    finally:
        del os.environ["T9_PASS2_BITS"]
