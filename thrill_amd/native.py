"""ctypes binding of libt9.so (include/thrill_amd.h).

Pointers are raw device addresses (torch tensor .data_ptr()); streams are
hipStream_t handles (torch.cuda.Stream.cuda_stream). No CPU fallback: a
missing library or GPU raises T9Error.
"""
import ctypes
import os

HERE = os.path.dirname(os.path.abspath(__file__))
SO = os.path.join(HERE, "libt9.so")

u64 = ctypes.c_uint64
u32 = ctypes.c_uint32
i32 = ctypes.c_int
vp = ctypes.c_void_p


class T9Error(RuntimeError):
    pass


def lib_path():
    return SO


def load_lib(path=None):
    path = path or SO
    if not os.path.exists(path):
        raise T9Error(
            f"libt9.so not found at {path}; build it with "
            "`make -C thrill_amd` (hipcc --offload-arch=gfx950). "
            "There is no CPU fallback.")
    return ctypes.CDLL(path)


_SIGS = {
    "t9_version": (ctypes.c_char_p, []),
    "t9_create": (i32, [ctypes.POINTER(vp), i32, i32, i32, vp]),
    "t9_destroy": (i32, [vp]),
    "t9_comm_id_size": (i32, []),
    "t9_comm_id": (i32, [vp]),
    "t9_comm_init": (i32, [vp, vp]),
    "t9_gen_u64": (i32, [vp, vp, u64, u64, u64, vp]),
    "t9_gen_records": (i32, [vp, vp, u64, u64, u64, vp]),
    "t9_sort_u64_workspace": (u64, [u64]),
    "t9_sort_u64": (i32, [vp, vp, u64, vp, vp]),
    "t9_sort_pairs_workspace": (u64, [u64]),
    "t9_sort_pairs_u64_u32": (i32, [vp, vp, vp, u64, vp, vp]),
    "t9_extract_key64": (i32, [vp, vp, u64, u32, u32, vp, vp, vp]),
    "t9_gather_records": (i32, [vp, vp, vp, u64, u32, vp, vp]),
    "t9_sort_records_workspace": (u64, [u64, u32]),
    "t9_sort_records": (i32, [vp, vp, vp, u64, u32, u32, vp, vp]),
    "t9_sort_records_keyle": (i32, [vp, vp, vp, u64, u32, vp, vp]),
    "t9_extract_key64_le": (i32, [vp, vp, u64, u32, u32, vp, vp, vp]),
    "t9_classify_u64": (i32, [vp, vp, u64, u64, vp, vp, u32, vp, vp, vp]),
    "t9_classify_rec": (i32, [vp, vp, vp, u64, u64, vp, vp, vp, u32, u32,
                              vp, vp, vp]),
    "t9_partition_idx_workspace": (u64, [u64]),
    "t9_partition_idx": (i32, [vp, vp, u64, u32, vp, vp, vp, vp]),
    "t9_alltoall": (i32, [vp, vp, vp, vp, vp, vp, vp, u64, vp]),
    "t9_hash_bucket": (i32, [vp, vp, u64, u64, u32, vp, vp, vp]),
    "t9_reduce_by_index": (i32, [vp, vp, vp, u64, u64, u64, vp, vp, vp]),
    "t9_index_bucket": (i32, [vp, vp, u64, u64, u64, u32, vp, vp, vp, vp]),
    "t9_reduce_init": (i32, [vp, vp, u64, vp]),
    "t9_reduce_build": (i32, [vp, vp, vp, u64, vp, u64, u64, vp, vp]),
    "t9_reduce_drain": (i32, [vp, vp, u64, vp, vp, vp, vp]),
    "t9_reduce128_init": (i32, [vp, vp, u64, vp]),
    "t9_reduce128_build": (i32, [vp, vp, vp, vp, u64, vp, u64, u64, vp,
                                 vp]),
    "t9_reduce128_drain": (i32, [vp, vp, u64, vp, vp, vp, vp, vp]),
    "t9_hash2_of": (i32, [vp, vp, u64, vp, vp, vp]),
    "t9_bucket_mod": (i32, [vp, vp, u64, u32, vp, vp, vp]),
    "t9_zipf_tokens": (i32, [vp, vp, vp, u64, u64, u64, u64, vp]),
    "t9_merge_u64": (i32, [vp, vp, u64, vp, u64, vp, vp]),
    "t9_merge_records": (i32, [vp, vp, u64, vp, u64, u32, vp, vp]),
    "t9_group_index_workspace": (u64, [u64]),
    "t9_group_index": (i32, [vp, vp, u64, vp, vp, vp, vp, vp]),
    "t9_perf_enable": (i32, [i32]),
    "t9_perf_read": (i32, [ctypes.c_char_p, ctypes.POINTER(ctypes.c_double),
                           ctypes.POINTER(u64)]),
    "t9_perf_reset": (i32, []),
}


class Native:
    """Loaded libt9 with a live t9_context. Requires a GPU."""

    def __init__(self, device=0, rank=0, world=1, comm=None):
        self._lib = load_lib()
        for name, (res, args) in _SIGS.items():
            fn = getattr(self._lib, name)
            fn.restype = res
            fn.argtypes = args
        ctx = vp()
        rc = self._lib.t9_create(ctypes.byref(ctx), device, rank, world,
                                 comm if comm else None)
        if rc != 0:
            raise T9Error(f"t9_create failed rc={rc} (no usable GPU?)")
        self.ctx = ctx

    def __getattr__(self, name):
        fn = getattr(self._lib, "t9_" + name, None)
        if fn is None:
            raise AttributeError(name)

        def call(*args):
            rc = fn(self.ctx, *args)
            if rc != 0:
                raise T9Error(f"t9_{name} failed rc={rc}")
            return rc
        return call

    # RCCL comm bootstrap (t9_comm_id takes no context argument)
    def comm_id(self):
        """rank 0: generate the ncclUniqueId as bytes."""
        size = self._lib.t9_comm_id_size()
        buf = (ctypes.c_uint8 * size)()
        rc = self._lib.t9_comm_id(buf)
        if rc != 0:
            raise T9Error(f"t9_comm_id failed rc={rc}")
        return bytes(buf)

    def comm_init(self, id_bytes):
        """every rank, collectively: connect the context's communicator."""
        buf = (ctypes.c_uint8 * len(id_bytes)).from_buffer_copy(id_bytes)
        rc = self._lib.t9_comm_init(self.ctx, buf)
        if rc != 0:
            raise T9Error(f"t9_comm_init failed rc={rc}")

    # perf registry functions take no context argument
    def perf_enable(self, on):
        self._lib.t9_perf_enable(1 if on else 0)

    def perf_read(self, cls):
        ms = ctypes.c_double()
        n = u64()
        rc = self._lib.t9_perf_read(cls.encode(), ctypes.byref(ms),
                                    ctypes.byref(n))
        if rc != 0:
            raise T9Error(f"t9_perf_read failed rc={rc}")
        return ms.value, n.value

    def perf_reset(self):
        self._lib.t9_perf_reset()

    def ws(self, name, *args):
        """workspace byte queries (no ctx argument)."""
        return getattr(self._lib, f"t9_{name}_workspace")(*args)

    def version(self):
        return self._lib.t9_version().decode()

    def close(self):
        if self.ctx:
            self._lib.t9_destroy(self.ctx)
            self.ctx = None
