"""thrill_amd — MI355X-native implementation of Thrill's Sort/ReduceByKey
DOp hot path. The compute path is hand-written HIP/CDNA4
kernels in libt9.so behind the C ABI of include/thrill_amd.h; this package
is the thin host layer (PyTorch supplies device memory, streams and
torch.distributed — plumbing, not the product).

There is NO CPU fallback anywhere in this package: if libt9.so or a GPU is
missing, operations raise. The CPU oracle lives in oracle/ and is test
infrastructure only.
"""
from .native import (  # noqa: F401
    T9Error,
    lib_path,
    load_lib,
    Native,
)
