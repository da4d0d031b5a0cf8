"""Product host orchestration of the TeraSort hot path:
the role of SortNode::MainOp (thrill/api/sort.hpp:537-663) — sample,
splitter selection on rank 0, classification, partition, all-to-all
exchange, local sort — with every bulk step a libt9 HIP kernel and the
BULK exchange through the C-ABI t9_alltoall (RCCL grouped send/recv over
xGMI, the product data plane; T9_EXCHANGE=torch restores the
torch.distributed all_to_all_single fallback). Host python here is
control plane only (splitters and counts are ~KBs, as in the reference
where FindAndSendSplitters runs on worker 0's CPU); scalar collectives go
through torch.distributed on whatever backend the process group uses —
gloo (CPU) by default in bench.py, so the C ABI owns ALL device traffic.

No CPU fallback: everything data-sized runs through the C ABI on the GPU.
"""
import ctypes
import math
import os

import numpy as np
import torch

from .native import Native

REC = 100  # TeraSort record bytes (examples/terasort/terasort.cpp:31-43)


def _ptr(t):
    return ctypes.c_void_p(t.data_ptr())


def _nptr(a):
    """host pointer of a (kept-alive) numpy array."""
    return ctypes.c_void_p(a.ctypes.data)


def _stream():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _ctl_device(dist):
    """device for small control-plane collectives: CPU under gloo (the
    bench default — keeps ALL device traffic inside the C ABI), CUDA
    under nccl."""
    return "cpu" if dist.get_backend() == "gloo" else "cuda"


def bootstrap_comm(nat, dist, rank):
    """Connect the context's RCCL communicator: rank 0 generates the
    ncclUniqueId (t9_comm_id), the id travels over the torch control
    plane (one broadcast at init), every rank joins via t9_comm_init —
    the same out-of-band endpoint exchange the reference does for its
    TCP mesh (api/context.cpp:604-614)."""
    obj = [nat.comm_id() if rank == 0 else None]
    dist.broadcast_object_list(obj, src=0)
    nat.comm_init(obj[0])


def exchange_counts(dist, send_counts, rank, world):
    """recv_counts[r] = what rank r sends me: all-gather the p x p count
    matrix over the control plane (tiny) and read my column. Replaces
    all_to_all_single(counts) which gloo does not support."""
    dev = _ctl_device(dist)
    mine = torch.from_numpy(np.ascontiguousarray(send_counts)).to(dev)
    rows = [torch.empty_like(mine) for _ in range(world)]
    dist.all_gather(rows, mine)
    return np.array([int(rows[r][rank].item()) for r in range(world)],
                    dtype=np.int64)


def displs_of(counts):
    d = np.zeros(len(counts), dtype=np.uint64)
    if len(counts) > 1:
        d[1:] = np.cumsum(counts[:-1].astype(np.uint64))
    return d


def a2a(nat, d_send, send_counts, d_recv, recv_counts, elem_size):
    """one all-to-all-v through the C ABI (t9_alltoall: RCCL grouped
    send/recv over xGMI; self-share via direct device copy; world==1
    shortcuts to a device memcpy). counts are element counts."""
    sc = np.ascontiguousarray(send_counts, dtype=np.uint64)
    rc = np.ascontiguousarray(recv_counts, dtype=np.uint64)
    sd = displs_of(sc)
    rd = displs_of(rc)
    nat.alltoall(_ptr(d_send), _nptr(sc), _nptr(sd), _ptr(d_recv),
                 _nptr(rc), _nptr(rd), elem_size, _stream())


def sample_size(n_total):
    """Reference sample-size law: log2(n) / eps^2, eps = 0.1
    (thrill/common/reservoir_sampling.hpp:269-274, api/sort.hpp:298)."""
    if n_total < 2:
        return 1
    return max(1, int(math.log2(n_total) * 100.0))


def select_splitters(sample_recs, sample_gidx, p):
    """FindAndSendSplitters (api/sort.hpp:337-378): sort the gathered
    samples and pick samples[(size_t)(i * size / p)]. Vectorized: sort by
    (u64 key prefix, global index) — classification is count-based
    (bucket = #{splitters < item}), so the partition is determined by the
    splitter SET, not their array order, and key-prefix ties among samples
    cannot affect correctness (they would only nudge bucket balance).
    sample_recs: (S, 100) uint8; returns (p-1, 100) records + (p-1,) idx."""
    k64 = sample_recs[:, :8].copy().view(">u8").reshape(-1).astype(np.uint64)
    order = np.lexsort((np.asarray(sample_gidx, dtype=np.uint64), k64))
    step = len(order) / p
    sel = order[[int(i * step) for i in range(1, p)]]
    return (sample_recs[sel],
            np.asarray(sample_gidx, dtype=np.uint64)[sel])


def k64_of_records(recs):
    """big-endian u64 prefix of each record (host-side, control plane)."""
    return np.array([int.from_bytes(r[:8].tobytes(), "big") for r in recs],
                    dtype=np.uint64)


def zipf_cdf(N, s, q=0.0):
    """CDF of the Zipf-Mandelbrot mass function 1/(H*(k+q)^s)
    (thrill/common/zipf_distribution.hpp:55-120). The canonical table for
    the product path; tests feed the SAME table to the oracle sampler, so
    parity is over the sampling given the table."""
    w = 1.0 / np.power(np.arange(1, N + 1, dtype=np.float64) + q, s)
    cdf = np.cumsum(w)
    cdf /= cdf[-1]
    return cdf


class WordCount:
    """One rank's ReduceByKey (word_count) state: Zipf tokens -> local
    pre-reduce (ReducePrePhase role, core/reduce_pre_phase.hpp) -> hash
    partition (core/reduce_functional.hpp:60-72) -> one all-to-all ->
    final reduce (post phase).

    keys128 (the default): words are dictionary-encoded into two
    independent 64-bit hashes and reduced on the 128-bit composite —
    string-identity semantics (the reference compares full keys,
    reduce_probing_hash_table.hpp:233); a forced single-hash collision
    keeps counts separate. keys128=False is the bare-u64 ReducePair
    path."""

    def __init__(self, n_total, vocab, s, seed, rank=0, world=1, device=0,
                 keys128=True):
        self.nat = Native(device=device, rank=rank, world=world)
        self.rank, self.world = rank, world
        if world > 1 and os.environ.get("T9_EXCHANGE", "t9") == "t9":
            import torch.distributed as dist
            bootstrap_comm(self.nat, dist, rank)
        self.n_total, self.seed = n_total, seed
        self.keys128 = keys128
        base = n_total // world
        rem = n_total % world
        self.n_local = base + (1 if rank < rem else 0)
        self.tok0 = rank * base + min(rank, rem)
        self.vocab = vocab
        cdf = zipf_cdf(vocab, s)
        self.d_cdf = torch.from_numpy(cdf).cuda()
        self.d_toks = torch.empty(self.n_local, dtype=torch.int64,
                                  device="cuda")
        self.d_ones = torch.ones(self.n_local, dtype=torch.int64,
                                 device="cuda")
        cap = 1 << max(10, int(math.ceil(math.log2(2 * vocab + 2))))
        self.cap = cap
        if keys128:
            self.d_k1 = torch.empty(self.n_local, dtype=torch.int64,
                                    device="cuda")
            self.d_k2 = torch.empty(self.n_local, dtype=torch.int64,
                                    device="cuda")
            # interleaved (k1, k2, sum) table: u64[3*cap]
            self.d_tbl = torch.empty(3 * cap, dtype=torch.int64,
                                     device="cuda")
            self.d_ok2 = torch.empty(cap, dtype=torch.int64, device="cuda")
        else:
            # interleaved (key, sum) table: u64[2*(cap+1)]
            self.d_tbl = torch.empty(2 * (cap + 1), dtype=torch.int64,
                                     device="cuda")
        self.d_ok = torch.empty(cap + 1, dtype=torch.int64, device="cuda")
        self.d_ov = torch.empty(cap + 1, dtype=torch.int64, device="cuda")
        self.d_err = torch.empty(1, dtype=torch.int32, device="cuda")
        self.d_n = torch.empty(1, dtype=torch.int64, device="cuda")

    def generate(self):
        self.nat.zipf_tokens(_ptr(self.d_toks), _ptr(self.d_cdf),
                             self.vocab, self.tok0, self.n_local,
                             self.seed, _stream())
        if self.keys128:
            self.nat.hash2_of(_ptr(self.d_toks), self.n_local,
                              _ptr(self.d_k1), _ptr(self.d_k2), _stream())

    def _reduce(self, d_keys, d_vals, n, salt=0):
        nat, s = self.nat, _stream()
        nat.reduce_init(_ptr(self.d_tbl), self.cap, s)
        nat.reduce_build(_ptr(d_keys), _ptr(d_vals), n, _ptr(self.d_tbl),
                         self.cap, salt, _ptr(self.d_err), s)
        nat.reduce_drain(_ptr(self.d_tbl), self.cap,
                         _ptr(self.d_ok), _ptr(self.d_ov), _ptr(self.d_n),
                         s)
        m = int(self.d_n.cpu().item())
        assert int(self.d_err.cpu().item()) == 0, "reduce table overflow"
        return self.d_ok[:m], self.d_ov[:m], m

    def _reduce128(self, d_k1, d_k2, d_vals, n, salt=0):
        """128-bit composite reduce; d_vals None = each pair counts 1."""
        nat, s = self.nat, _stream()
        nat.reduce128_init(_ptr(self.d_tbl), self.cap, s)
        nat.reduce128_build(_ptr(d_k1), _ptr(d_k2),
                            _ptr(d_vals) if d_vals is not None else None,
                            n, _ptr(self.d_tbl), self.cap, salt,
                            _ptr(self.d_err), s)
        nat.reduce128_drain(_ptr(self.d_tbl), self.cap, _ptr(self.d_ok),
                            _ptr(self.d_ok2), _ptr(self.d_ov),
                            _ptr(self.d_n), s)
        m = int(self.d_n.cpu().item())
        assert int(self.d_err.cpu().item()) == 0, "reduce table overflow"
        return self.d_ok[:m], self.d_ok2[:m], self.d_ov[:m], m

    def _exchange(self, arrays, send_counts, recv_counts, dist):
        """all-to-all-v each u64 array with the same counts; returns the
        received arrays."""
        n_recv = int(recv_counts.sum())
        outs = []
        exchange = os.environ.get("T9_EXCHANGE", "t9")
        for a in arrays:
            r = torch.empty(max(n_recv, 1), dtype=torch.int64,
                            device="cuda")
            if exchange == "t9" or self.world == 1:
                a2a(self.nat, a, send_counts, r, recv_counts, 8)
            else:
                dist.all_to_all_single(
                    r[:n_recv], a[:int(send_counts.sum())],
                    output_split_sizes=recv_counts.tolist(),
                    input_split_sizes=send_counts.tolist())
            outs.append(r)
        return outs, n_recv

    def step(self):
        """One full ReduceByKey of the (distributed) token stream.
        Returns (keys tensor, vals tensor, m) of this rank's final pairs
        (keys128: (k1 tensor, k2 tensor, vals tensor, m))."""
        nat, s = self.nat, _stream()
        if self.keys128:
            return self._step128()
        ok, ov, m = self._reduce(self.d_toks, self.d_ones, self.n_local)
        if self.world == 1 and not os.environ.get("T9_FORCE_DIST"):
            return ok.clone(), ov.clone(), m

        import torch.distributed as dist
        p = self.world
        keys = ok.clone()
        vals = ov.clone()
        d_bucket = torch.empty(max(m, 1), dtype=torch.int32, device="cuda")
        d_counts = torch.empty(p, dtype=torch.int64, device="cuda")
        nat.hash_bucket(_ptr(keys), m, 0, p, _ptr(d_bucket),
                        _ptr(d_counts), s)
        d_perm = torch.empty(max(m, 1), dtype=torch.int32, device="cuda")
        d_offs = torch.empty(p + 1, dtype=torch.int64, device="cuda")
        d_ws = torch.empty(max(int(self.nat.ws("partition_idx", m)), 256),
                           dtype=torch.uint8, device="cuda")
        nat.partition_idx(_ptr(d_bucket), m, p, _ptr(d_perm), _ptr(d_offs),
                          _ptr(d_ws), s)
        ks = torch.empty_like(keys)
        vs = torch.empty_like(vals)
        if m:
            nat.gather_records(_ptr(keys), _ptr(d_perm), m, 8, _ptr(ks), s)
            nat.gather_records(_ptr(vals), _ptr(d_perm), m, 8, _ptr(vs), s)
        send_counts = d_counts.cpu().numpy().astype(np.int64)
        recv_counts = exchange_counts(dist, send_counts, self.rank,
                                      self.world)
        (rk, rv), n_recv = self._exchange([ks, vs], send_counts,
                                          recv_counts, dist)
        ok2, ov2, m2 = self._reduce(rk, rv, n_recv)
        return ok2.clone(), ov2.clone(), m2

    def _step128(self):
        nat, s = self.nat, _stream()
        k1, k2, v, m = self._reduce128(self.d_k1, self.d_k2, None,
                                       self.n_local)
        if self.world == 1 and not os.environ.get("T9_FORCE_DIST"):
            return k1.clone(), k2.clone(), v.clone(), m

        import torch.distributed as dist
        p = self.world
        k1, k2, v = k1.clone(), k2.clone(), v.clone()
        d_bucket = torch.empty(max(m, 1), dtype=torch.int32, device="cuda")
        d_counts = torch.empty(p, dtype=torch.int64, device="cuda")
        # partition on k1 (already the hash): bucket = k1 % p
        nat.bucket_mod(_ptr(k1), m, p, _ptr(d_bucket), _ptr(d_counts), s)
        d_perm = torch.empty(max(m, 1), dtype=torch.int32, device="cuda")
        d_offs = torch.empty(p + 1, dtype=torch.int64, device="cuda")
        d_ws = torch.empty(max(int(self.nat.ws("partition_idx", m)), 256),
                           dtype=torch.uint8, device="cuda")
        nat.partition_idx(_ptr(d_bucket), m, p, _ptr(d_perm), _ptr(d_offs),
                          _ptr(d_ws), s)
        g1 = torch.empty_like(k1)
        g2 = torch.empty_like(k2)
        gv = torch.empty_like(v)
        if m:
            nat.gather_records(_ptr(k1), _ptr(d_perm), m, 8, _ptr(g1), s)
            nat.gather_records(_ptr(k2), _ptr(d_perm), m, 8, _ptr(g2), s)
            nat.gather_records(_ptr(v), _ptr(d_perm), m, 8, _ptr(gv), s)
        send_counts = d_counts.cpu().numpy().astype(np.int64)
        recv_counts = exchange_counts(dist, send_counts, self.rank,
                                      self.world)
        (r1, r2, rv), n_recv = self._exchange([g1, g2, gv], send_counts,
                                              recv_counts, dist)
        o1, o2, ov, m2 = self._reduce128(r1, r2, rv, n_recv)
        return o1.clone(), o2.clone(), ov.clone(), m2

    def close(self):
        self.nat.close()


class TeraSort:
    """One rank's TeraSort state. world==1: pure local sort. world>1:
    sample -> splitters -> classify -> partition -> all-to-all -> local
    sort; output = this rank's globally-contiguous sorted shard."""

    def __init__(self, n_total, seed, rank=0, world=1, device=0):
        self.nat = Native(device=device, rank=rank, world=world)
        self.rank, self.world = rank, world
        self.n_total, self.seed = n_total, seed
        self.exchange = os.environ.get("T9_EXCHANGE", "t9")
        if world > 1 and self.exchange == "t9":
            import torch.distributed as dist
            bootstrap_comm(self.nat, dist, rank)
        base = n_total // world
        rem = n_total % world
        self.n_local = base + (1 if rank < rem else 0)
        self.gidx0 = rank * base + min(rank, rem)
        self.d_in = torch.empty(self.n_local * REC, dtype=torch.uint8,
                                device="cuda")
        self.d_out = torch.empty(self.n_local * REC, dtype=torch.uint8,
                                 device="cuda")
        ws_bytes = max(
            self.nat.ws("sort_records", self.n_local, REC),
            self.nat.ws("partition_idx", self.n_local),
        )
        if world > 1:
            # headroom for sorting a received shard up to ~1.3x the even
            # share (splitter imbalance target eps=0.1, api/sort.hpp:298)
            ws_bytes = max(ws_bytes, self.nat.ws(
                "sort_records", self.n_local + self.n_local // 3 + 16, REC))
        self.d_ws = torch.empty(int(ws_bytes), dtype=torch.uint8,
                                device="cuda")
        if world > 1 or os.environ.get("T9_FORCE_DIST"):
            self.d_keys = torch.empty(self.n_local, dtype=torch.int64,
                                      device="cuda")
            self.d_idx = torch.empty(self.n_local, dtype=torch.int32,
                                     device="cuda")
            self.d_bucket = torch.empty(self.n_local, dtype=torch.int32,
                                        device="cuda")
            self.d_counts = torch.empty(world, dtype=torch.int64,
                                        device="cuda")
            self.d_perm = torch.empty(self.n_local, dtype=torch.int32,
                                      device="cuda")
            self.d_send = torch.empty(self.n_local * REC, dtype=torch.uint8,
                                      device="cuda")

    def generate(self):
        self.nat.gen_records(_ptr(self.d_in), self.gidx0, self.n_local,
                             self.seed, _stream())

    def _splitters(self):
        """sample locally, gather to rank 0, select, broadcast. Returns
        device tensors (spl_recs bytes, spl_k64, spl_idx)."""
        import torch.distributed as dist
        if self.world == 1:
            # forced-distributed self-exchange: no splitters needed
            z = torch.zeros(REC, dtype=torch.uint8, device="cuda")
            zk = torch.zeros(1, dtype=torch.int64, device="cuda")
            zi = torch.zeros(1, dtype=torch.int64, device="cuda")
            return z, zk, zi
        # S is derived identically on every rank with no communication:
        # capped by the minimum shard size (= base share), so the sample
        # tensors are equal-sized and plain all_gather works (no object
        # collectives — they cost milliseconds per step at N=8).
        base = self.n_total // self.world
        S = max(1, min(max(base, 1), sample_size(self.n_total) // self.world))
        stride = max(1, self.n_local // S)
        pos = torch.arange(0, self.n_local, stride, device="cuda")[:S]
        if len(pos) < S:   # only possible for tiny shards
            # degenerate shard (n_local < S, possibly 0): pad with repeats
            # of the last sample, or position 0 for an empty shard — the
            # splitter rule is invariant to duplicate samples
            pad = pos[-1:] if len(pos) else torch.zeros(
                1, dtype=pos.dtype, device=pos.device)
            pos = torch.cat([pos, pad.expand(S - len(pos))])
        dpos = pos.to(torch.int32)
        d_samp = torch.zeros(S * REC, dtype=torch.uint8, device="cuda")
        if self.n_local:
            self.nat.gather_records(_ptr(self.d_in), _ptr(dpos), S, REC,
                                    _ptr(d_samp), _stream())
        gidx = (pos + self.gidx0).to(torch.int64)
        # samples are ~KBs: gather + splitter broadcast run on the control
        # plane (CPU tensors under gloo, CUDA under nccl)
        ctl = _ctl_device(dist)
        samp_c = d_samp.to(ctl)
        gidx_c = gidx.to(ctl)
        gs_l = [torch.empty_like(samp_c) for _ in range(self.world)]
        gi_l = [torch.empty_like(gidx_c) for _ in range(self.world)]
        dist.all_gather(gs_l, samp_c)
        dist.all_gather(gi_l, gidx_c)
        p = self.world
        spl_recs_c = torch.empty(max(p - 1, 1) * REC, dtype=torch.uint8,
                                 device=ctl)
        spl_idx_c = torch.empty(max(p - 1, 1), dtype=torch.int64,
                                device=ctl)
        if self.rank == 0:
            all_recs = torch.cat(gs_l).cpu().numpy().reshape(
                self.world * S, REC)
            all_idx = torch.cat(gi_l).cpu().numpy().astype(np.uint64)
            spl_recs, spl_idx = select_splitters(all_recs, all_idx, p)
            spl_recs_c.copy_(torch.from_numpy(
                spl_recs.reshape(-1).copy()).to(ctl))
            spl_idx_c.copy_(torch.from_numpy(
                spl_idx.view(np.int64).copy()).to(ctl))
        dist.broadcast(spl_recs_c, 0)
        dist.broadcast(spl_idx_c, 0)
        spl_recs_t = spl_recs_c.cuda()
        spl_idx_t = spl_idx_c.cuda()
        spl_recs = spl_recs_t.cpu().numpy().reshape(p - 1, REC)
        spl_k64 = torch.from_numpy(
            k64_of_records(spl_recs).view(np.int64)).cuda()
        return spl_recs_t, spl_k64, spl_idx_t

    def step(self):
        """One full TeraSort of the (distributed) input. Returns the local
        output tensor (n_out*100 bytes) and n_out.

        T9_FORCE_DIST=1 routes world==1 through the distributed branch
        (self-exchange) so the exact multi-rank code path is testable on
        one GPU."""
        nat, s = self.nat, _stream()
        if self.world == 1 and not os.environ.get("T9_FORCE_DIST"):
            nat.sort_records(_ptr(self.d_in), _ptr(self.d_out),
                             self.n_local, REC, 10, _ptr(self.d_ws), s)
            return self.d_out, self.n_local

        import torch.distributed as dist
        nat.extract_key64(_ptr(self.d_in), self.n_local, REC, 0,
                          _ptr(self.d_keys), _ptr(self.d_idx), s)
        spl_recs_t, spl_k64, spl_idx_t = self._splitters()
        p = self.world
        nat.classify_rec(_ptr(self.d_in), _ptr(self.d_keys), self.n_local,
                         self.gidx0, _ptr(spl_recs_t), _ptr(spl_k64),
                         _ptr(spl_idx_t), p, REC, _ptr(self.d_bucket),
                         _ptr(self.d_counts), s)
        d_offs = torch.empty(p + 1, dtype=torch.int64, device="cuda")
        nat.partition_idx(_ptr(self.d_bucket), self.n_local, p,
                          _ptr(self.d_perm), _ptr(d_offs), _ptr(self.d_ws),
                          s)
        nat.gather_records(_ptr(self.d_in), _ptr(self.d_perm), self.n_local,
                           REC, _ptr(self.d_send), s)
        send_counts = self.d_counts.cpu().numpy().astype(np.int64)
        recv_counts = exchange_counts(dist, send_counts, self.rank,
                                      self.world)
        n_recv = int(recv_counts.sum())
        d_recv = torch.empty(max(n_recv, 1) * REC, dtype=torch.uint8,
                             device="cuda")
        if self.exchange == "t9" or self.world == 1:
            # the product exchange: C-ABI RCCL all-to-all-v over xGMI
            # (t9_alltoall; self-share via direct device copy, world==1
            # shortcuts entirely to a device memcpy)
            a2a(nat, self.d_send, send_counts, d_recv, recv_counts, REC)
        else:
            dist.all_to_all_single(
                d_recv[:n_recv * REC], self.d_send,
                output_split_sizes=(recv_counts * REC).tolist(),
                input_split_sizes=(send_counts * REC).tolist())
        d_sorted = torch.empty(max(n_recv, 1) * REC, dtype=torch.uint8,
                               device="cuda")
        ws_need = nat.ws("sort_records", n_recv, REC)
        ws = self.d_ws if ws_need <= self.d_ws.numel() \
            else torch.empty(int(ws_need), dtype=torch.uint8, device="cuda")
        nat.sort_records(_ptr(d_recv), _ptr(d_sorted), n_recv, REC, 10,
                         _ptr(ws), s)
        return d_sorted, n_recv

    def close(self):
        self.nat.close()
