"""ctypes wrapper around oracle/liboracle_t9.so.

TEST INFRASTRUCTURE ONLY (plus bench.py's cpu_baseline leg). The product
path (thrill_amd/) never imports this module.
"""
import ctypes
import os

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(REPO, "oracle", "liboracle_t9.so")

u64p = np.ctypeslib.ndpointer(dtype=np.uint64, flags="C_CONTIGUOUS")
u32p = np.ctypeslib.ndpointer(dtype=np.uint32, flags="C_CONTIGUOUS")
u8p = np.ctypeslib.ndpointer(dtype=np.uint8, flags="C_CONTIGUOUS")
f64p = np.ctypeslib.ndpointer(dtype=np.float64, flags="C_CONTIGUOUS")
u64 = ctypes.c_uint64
u32 = ctypes.c_uint32
f64 = ctypes.c_double


class Oracle:
    def __init__(self, path=SO):
        lib = ctypes.CDLL(path)
        self._lib = lib
        lib.t9o_hash128to64.restype = u64
        lib.t9o_hash128to64.argtypes = [u64, u64]
        lib.t9o_partition_of_u64.restype = u32
        lib.t9o_partition_of_u64.argtypes = [u64, u64, u32]
        lib.t9o_splitmix64_at.restype = u64
        lib.t9o_splitmix64_at.argtypes = [u64, u64]
        lib.t9o_gen_u64.argtypes = [u64p, u64, u64, u64]
        lib.t9o_gen_records.argtypes = [u8p, u64, u64, u64]
        lib.t9o_sort_u64.argtypes = [u64p, u64]
        lib.t9o_sort_records.argtypes = [u8p, u64, u32]
        lib.t9o_sort_records_parallel.restype = ctypes.c_int
        lib.t9o_sort_records_parallel.argtypes = [u8p, u64, u32]
        lib.t9o_select_splitters_u64.argtypes = [
            u64p, u64p, u64, u32, u64p, u64p]
        lib.t9o_classify_u64.argtypes = [u64p, u64, u64, u64p, u64p, u32, u32p]
        lib.t9o_classify_u64_tree.argtypes = [
            u64p, u64, u64, u64p, u64p, u32, u32p]
        lib.t9o_classify_rec.argtypes = [
            u8p, u64, u64, u32, u32, u8p, u64p, u32, u32p]
        lib.t9o_reduce_u64.restype = u64
        lib.t9o_reduce_u64.argtypes = [
            u64p, u64p, u64, u64, u64, u64p, u64p, u64]
        lib.t9o_reduce_by_index.restype = ctypes.c_int
        lib.t9o_reduce_by_index.argtypes = [u64p, u64p, u64, u64, u64, u64p]
        lib.t9o_zipf_cdf.argtypes = [f64p, u64, f64, f64]
        lib.t9o_zipf_tokens.argtypes = [u64p, f64p, u64, u64, u64, u64]

    def hash128to64(self, upper, lower):
        return self._lib.t9o_hash128to64(upper, lower)

    def partition_of_u64(self, key, salt, p):
        return self._lib.t9o_partition_of_u64(key, salt, p)

    def gen_u64(self, n, seed, index0=0):
        out = np.empty(n, dtype=np.uint64)
        self._lib.t9o_gen_u64(out, index0, n, seed)
        return out

    def gen_records(self, n, seed, index0=0):
        out = np.empty(n * 100, dtype=np.uint8)
        self._lib.t9o_gen_records(out, index0, n, seed)
        return out.reshape(n, 100)

    def sort_u64(self, keys):
        keys = np.ascontiguousarray(keys, dtype=np.uint64).copy()
        self._lib.t9o_sort_u64(keys, len(keys))
        return keys

    def sort_records(self, recs):
        recs = np.ascontiguousarray(recs, dtype=np.uint8).copy()
        n, rec_size = recs.shape
        self._lib.t9o_sort_records(recs.reshape(-1), n, rec_size)
        return recs.reshape(n, rec_size)

    def sort_records_parallel(self, recs):
        recs = np.ascontiguousarray(recs, dtype=np.uint8).copy()
        n, rec_size = recs.shape
        threads = self._lib.t9o_sort_records_parallel(
            recs.reshape(-1), n, rec_size)
        return recs.reshape(n, rec_size), threads

    def select_splitters_u64(self, sample_keys, sample_idx, p):
        sample_keys = np.ascontiguousarray(sample_keys, dtype=np.uint64)
        sample_idx = np.ascontiguousarray(sample_idx, dtype=np.uint64)
        ok = np.empty(p - 1, dtype=np.uint64)
        oi = np.empty(p - 1, dtype=np.uint64)
        self._lib.t9o_select_splitters_u64(
            sample_keys, sample_idx, len(sample_keys), p, ok, oi)
        return ok, oi

    def classify_u64(self, keys, gidx0, spl_keys, spl_idx, p, tree=False):
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        spl_keys = np.ascontiguousarray(spl_keys, dtype=np.uint64)
        spl_idx = np.ascontiguousarray(spl_idx, dtype=np.uint64)
        out = np.empty(len(keys), dtype=np.uint32)
        fn = (self._lib.t9o_classify_u64_tree if tree
              else self._lib.t9o_classify_u64)
        fn(keys, len(keys), gidx0, spl_keys, spl_idx, p, out)
        return out

    def classify_rec(self, recs, gidx0, key_len, splitters, spl_idx, p):
        recs = np.ascontiguousarray(recs, dtype=np.uint8)
        n, rec_size = recs.shape
        splitters = np.ascontiguousarray(splitters, dtype=np.uint8)
        spl_idx = np.ascontiguousarray(spl_idx, dtype=np.uint64)
        out = np.empty(n, dtype=np.uint32)
        self._lib.t9o_classify_rec(
            recs.reshape(-1), n, gidx0, rec_size, key_len,
            splitters.reshape(-1), spl_idx, p, out)
        return out

    def reduce128(self, k1, k2, vals=None, salt=0, cap=None):
        """128-bit composite-key reduce restatement (oracle/t9_oracle.cpp
        t9o_reduce128): probing-table semantics with equality on the
        (k1, k2) pair; output sorted by (k1, k2)."""
        import ctypes
        import numpy as np
        k1 = np.ascontiguousarray(k1, dtype=np.uint64)
        k2 = np.ascontiguousarray(k2, dtype=np.uint64)
        n = len(k1)
        if vals is not None:
            vals = np.ascontiguousarray(vals, dtype=np.uint64)
        cap = cap or max(16, 2 * n)
        o1 = np.empty(cap, np.uint64)
        o2 = np.empty(cap, np.uint64)
        ov = np.empty(cap, np.uint64)
        fn = self._lib.t9o_reduce128
        fn.restype = ctypes.c_uint64
        m = fn(k1.ctypes.data_as(ctypes.c_void_p),
               k2.ctypes.data_as(ctypes.c_void_p),
               vals.ctypes.data_as(ctypes.c_void_p)
               if vals is not None else None,
               ctypes.c_uint64(n), ctypes.c_uint64(salt),
               o1.ctypes.data_as(ctypes.c_void_p),
               o2.ctypes.data_as(ctypes.c_void_p),
               ov.ctypes.data_as(ctypes.c_void_p),
               ctypes.c_uint64(cap))
        assert m != 2**64 - 1, "oracle reduce128 cap overflow"
        m = int(m)
        return o1[:m], o2[:m], ov[:m]

    def reduce_u64(self, keys, vals, salt=0, num_partitions=1, cap=None):
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        vals = np.ascontiguousarray(vals, dtype=np.uint64)
        if cap is None:
            cap = len(keys) + 1
        ok = np.empty(cap, dtype=np.uint64)
        ov = np.empty(cap, dtype=np.uint64)
        m = self._lib.t9o_reduce_u64(
            keys, vals, len(keys), salt, num_partitions, ok, ov, cap)
        assert m != np.iinfo(np.uint64).max, "oracle reduce: cap exceeded"
        return ok[:m].copy(), ov[:m].copy()

    def reduce_by_index(self, keys, vals, begin, size):
        keys = np.ascontiguousarray(keys, dtype=np.uint64)
        vals = np.ascontiguousarray(vals, dtype=np.uint64)
        dense = np.empty(size, dtype=np.uint64)
        rc = self._lib.t9o_reduce_by_index(keys, vals, len(keys), begin,
                                           size, dense)
        assert rc == 0, "key out of range"
        return dense

    def zipf_cdf(self, N, s, q=0.0):
        cdf = np.empty(N, dtype=np.float64)
        self._lib.t9o_zipf_cdf(cdf, N, s, q)
        return cdf

    def zipf_tokens(self, cdf, n, seed, index0=0):
        out = np.empty(n, dtype=np.uint64)
        self._lib.t9o_zipf_tokens(out, cdf, len(cdf), index0, n, seed)
        return out
