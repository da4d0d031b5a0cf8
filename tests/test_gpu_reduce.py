"""GPU parity tests for the reduce path (ReduceByKey semantics):
device open-addressing table vs the oracle's probing-table restatement.
Parity = key-sorted (key,value) multiset equality, bit-exact (u64 sums).
"""
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


def run_reduce(nat, keys, vals, cap, salt=0):
    n = len(keys)
    dk, dv = G.dev(keys), G.dev(vals)
    tbl = G.empty(2 * (cap + 1), np.uint64)
    derr = G.empty(1, np.uint32)
    dn = G.empty(1, np.uint64)
    ok, ov = G.empty(cap + 1, np.uint64), G.empty(cap + 1, np.uint64)
    s = G.stream()
    nat.reduce_init(G.ptr(tbl), cap, s)
    nat.reduce_build(G.ptr(dk), G.ptr(dv), n, G.ptr(tbl), cap,
                     salt, G.ptr(derr), s)
    nat.reduce_drain(G.ptr(tbl), cap, G.ptr(ok), G.ptr(ov),
                     G.ptr(dn), s)
    assert int(G.host(derr, np.uint32)[0]) == 0, "table overflow"
    m = int(G.host(dn, np.uint64)[0])
    gk = G.host(ok, np.uint64)[:m]
    gv = G.host(ov, np.uint64)[:m]
    order = np.argsort(gk)
    return gk[order], gv[order]


def test_reduce_modulo_sums(nat, oracle):
    # reduce_node_test.cpp:83-137 pattern
    n, mod = 1 << 20, 601   # 601 keys x occurrences: reduce_pre_phase_test
    keys = (np.arange(n, dtype=np.uint64) % mod).astype(np.uint64)
    vals = np.ones(n, dtype=np.uint64)
    gk, gv = run_reduce(nat, keys, vals, cap=4096)
    ek, ev = oracle.reduce_u64(keys, vals)
    assert np.array_equal(gk, ek)
    assert np.array_equal(gv, ev)


def test_reduce_random_values(nat, oracle):
    rng = np.random.default_rng(8)
    n = 1 << 19
    keys = rng.integers(0, 10_000, n).astype(np.uint64)
    vals = rng.integers(0, 1 << 40, n).astype(np.uint64)
    gk, gv = run_reduce(nat, keys, vals, cap=1 << 15)
    ek, ev = oracle.reduce_u64(keys, vals)
    assert np.array_equal(gk, ek) and np.array_equal(gv, ev)


def test_reduce_sentinel_and_extreme_keys(nat, oracle):
    # key 0xFFFF..F is the table's empty sentinel — dedicated-slot path
    # (mirrors reduce_probing_hash_table.hpp:195-217); key 0 is the
    # reference's own sentinel.
    keys = np.array([0, 2**64 - 1, 5, 2**64 - 1, 0, 5, 2**64 - 1],
                    dtype=np.uint64)
    vals = np.array([1, 10, 100, 20, 2, 200, 30], dtype=np.uint64)
    gk, gv = run_reduce(nat, keys, vals, cap=16)
    ek, ev = oracle.reduce_u64(keys, vals)
    assert np.array_equal(gk, ek) and np.array_equal(gv, ev)


def test_reduce_zipf_skew(nat, oracle):
    # Zipf(1.1) heavy head: exercises the wave-combine skew control.
    N = 10_000
    cdf = oracle.zipf_cdf(N, 1.1)
    toks = oracle.zipf_tokens(cdf, 1 << 20, seed=13)
    vals = np.ones(len(toks), dtype=np.uint64)
    gk, gv = run_reduce(nat, toks, vals, cap=1 << 15)
    ek, ev = oracle.reduce_u64(toks, vals)
    assert np.array_equal(gk, ek) and np.array_equal(gv, ev)


def test_reduce_empty(nat):
    gk, gv = run_reduce(nat, np.empty(0, np.uint64), np.empty(0, np.uint64),
                        cap=16)
    assert len(gk) == 0


def test_zipf_tokens_gpu_matches_oracle(nat, oracle):
    N = 100_000
    cdf = oracle.zipf_cdf(N, 1.1)
    n = 1 << 20
    dcdf = G.dev(cdf)
    dout = G.empty(n, np.uint64)
    nat.zipf_tokens(G.ptr(dout), G.ptr(dcdf), N, 0, n, 77, G.stream())
    got = G.host(dout, np.uint64)
    assert np.array_equal(got, oracle.zipf_tokens(cdf, n, seed=77))


def test_bacon_ipsum_kat_gpu(nat, oracle):
    """The reference's one true in-repo KAT (word_count_test.cpp:36-79)
    through the GPU reduce path."""
    import json
    import os
    here = os.path.dirname(os.path.abspath(__file__))
    with open(os.path.join(here, "golden", "bacon_ipsum_correct.json")) as f:
        table = json.load(f)
    words = []
    with open(os.path.join(here, "golden", "wordcount.in")) as f:
        for line in f:
            words += [w for w in line.rstrip("\n").split(" ") if w]
    vocab = sorted(set(words))
    wid = {w: oracle.hash128to64(1, i) for i, w in enumerate(vocab)}
    keys = np.array([wid[w] for w in words], dtype=np.uint64)
    vals = np.ones(len(words), dtype=np.uint64)
    gk, gv = run_reduce(nat, keys, vals, cap=256)
    back = {h: w for w, h in wid.items()}
    result = {back[int(k)]: int(v) for k, v in zip(gk, gv)}
    assert result == table


def test_hash_bucket_matches_reference_mapping(nat, oracle):
    n, p = 200_000, 8
    keys = oracle.gen_u64(n, seed=31)
    dk = G.dev(keys)
    db = G.empty(n, np.uint32)
    dc = G.empty(p, np.uint64)
    nat.hash_bucket(G.ptr(dk), n, 0, p, G.ptr(db), G.ptr(dc), G.stream())
    got = G.host(db, np.uint32)
    expect = np.array([oracle.partition_of_u64(int(k), 0, p)
                       for k in keys[:2000]], dtype=np.uint32)
    assert np.array_equal(got[:2000], expect)
    counts = G.host(dc, np.uint64)
    assert counts.sum() == n


def test_wordcount_pipeline_single_gpu(nat, oracle):
    from thrill_amd.pipeline import WordCount, zipf_cdf
    n, vocab = 1 << 20, 50_000
    wc = WordCount(n, vocab, 1.1, seed=5, rank=0, world=1, device=0,
                   keys128=False)
    wc.generate()
    ok, ov, m = wc.step()
    gk = G.host(ok, np.uint64)
    gv = G.host(ov, np.uint64)
    order = np.argsort(gk)
    # oracle on the SAME cdf table -> identical token stream
    toks = oracle.zipf_tokens(zipf_cdf(vocab, 1.1), n, seed=5)
    ek, ev = oracle.reduce_u64(toks, np.ones(n, np.uint64))
    assert np.array_equal(gk[order], ek)
    assert np.array_equal(gv[order], ev)
    wc.close()


def test_reduce_by_index_parity(nat, oracle):
    # ReduceToIndex: dense per-index u64 sums.
    rng = np.random.default_rng(17)
    n, begin, size = 1 << 20, 1000, 5000
    keys = rng.integers(begin, begin + size, n).astype(np.uint64)
    vals = rng.integers(0, 1 << 40, n).astype(np.uint64)
    dk, dv = G.dev(keys), G.dev(vals)
    dd = G.empty(size, np.uint64)
    de = G.empty(1, np.uint32)
    nat.reduce_by_index(G.ptr(dk), G.ptr(dv), n, begin, size, G.ptr(dd),
                        G.ptr(de), G.stream())
    assert int(G.host(de, np.uint32)[0]) == 0
    got = G.host(dd, np.uint64)
    assert np.array_equal(got, oracle.reduce_by_index(keys, vals, begin,
                                                      size))


def test_index_bucket_mapping(nat):
    # bucket = (k-begin)*p/size (core/reduce_functional.hpp:113-128)
    n, begin, size, p = 100_000, 50, 1024, 8
    rng = np.random.default_rng(18)
    keys = rng.integers(begin, begin + size, n).astype(np.uint64)
    dk = G.dev(keys)
    db = G.empty(n, np.uint32)
    dc = G.empty(p, np.uint64)
    de = G.empty(1, np.uint32)
    nat.index_bucket(G.ptr(dk), n, begin, size, p, G.ptr(db), G.ptr(dc),
                     G.ptr(de), G.stream())
    got = G.host(db, np.uint32)
    expect = ((keys - begin) * p // size).astype(np.uint32)
    assert np.array_equal(got, expect)
    assert int(G.host(de, np.uint32)[0]) == 0

    # out-of-range key: clamps to the last partition AND flags the error
    keys2 = keys.copy()
    keys2[123] = begin + size + 7
    dk2 = G.dev(keys2)
    nat.index_bucket(G.ptr(dk2), n, begin, size, p, G.ptr(db), G.ptr(dc),
                     G.ptr(de), G.stream())
    got2 = G.host(db, np.uint32)
    assert got2[123] == p - 1
    assert int(G.host(de, np.uint32)[0]) == 1


def test_reduce_overflow_sets_error(nat):
    # more distinct keys than the table can hold: d_error must be set and
    # the kernel must terminate (bounded probe loop), not hang.
    n, cap = 10_000, 1024
    keys = np.arange(1, n + 1, dtype=np.uint64)
    vals = np.ones(n, dtype=np.uint64)
    dk, dv = G.dev(keys), G.dev(vals)
    tbl = G.empty(2 * (cap + 1), np.uint64)
    derr = G.empty(1, np.uint32)
    s = G.stream()
    nat.reduce_init(G.ptr(tbl), cap, s)
    nat.reduce_build(G.ptr(dk), G.ptr(dv), n, G.ptr(tbl), cap,
                     0, G.ptr(derr), s)
    assert int(G.host(derr, np.uint32)[0]) == 1


def test_group_index_parity(nat):
    # group index over sorted keys vs numpy unique run starts
    rng = np.random.default_rng(23)
    keys = np.sort(rng.integers(0, 5000, 1 << 20).astype(np.uint64))
    dk = G.dev(keys)
    du = G.empty(len(keys), np.uint64)
    do = G.empty(len(keys), np.uint64)
    dc = G.empty(1, np.uint64)
    w = G.ws(nat.ws("group_index", len(keys)))
    nat.group_index(G.ptr(dk), len(keys), G.ptr(du), G.ptr(do), G.ptr(dc),
                    G.ptr(w), G.stream())
    m = int(G.host(dc, np.uint64)[0])
    uk, idx = np.unique(keys, return_index=True)
    assert m == len(uk)
    assert np.array_equal(G.host(du, np.uint64)[:m], uk)
    assert np.array_equal(G.host(do, np.uint64)[:m], idx.astype(np.uint64))
