"""GPU-test helpers: torch supplies device memory and streams (plumbing);
all compute goes through the libt9 C ABI (the product path)."""
import ctypes

import numpy as np
import torch


def dev(np_arr):
    """numpy -> device tensor preserving bytes (u64 -> int64 view, etc.)."""
    a = np.ascontiguousarray(np_arr)
    if a.dtype == np.uint64:
        t = torch.from_numpy(a.view(np.int64))
    elif a.dtype == np.uint32:
        t = torch.from_numpy(a.view(np.int32))
    else:
        t = torch.from_numpy(a)
    return t.cuda()


def host(tensor, np_dtype):
    a = tensor.cpu().numpy()
    return a.view(np_dtype)


def empty(n, np_dtype):
    m = {np.uint64: torch.int64, np.uint32: torch.int32,
         np.uint8: torch.uint8, np.float64: torch.float64}
    return torch.empty(int(n), dtype=m[np_dtype], device="cuda")


def ws(nbytes):
    return torch.empty(max(int(nbytes), 256), dtype=torch.uint8,
                       device="cuda")


def ptr(t):
    return ctypes.c_void_p(t.data_ptr())


def stream():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
