"""gloo_amd: an MI355X-native collective communications framework.

Brand-new implementation with the capability surface of pytorch/gloo
(see SURVEY.md): Context / transport Pair / rendezvous Store public API,
CPU collectives over an epoll TCP transport, and HIP/CDNA4 device
collectives over xGMI (hipIpcMemHandle peer transport) for GPU tensors.
"""

import os as _os

# 8 hardware queues per process so each pooled stream owns one (the
# ROCm default of 4 would multiplex streams onto shared queues, where a
# doorbell spin kernel head-of-line-blocks the stream behind it).
_os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")

# Load torch (and its bundled HIP runtime) BEFORE our extension: _C's
# libamdhip64.so.7 dependency then resolves by SONAME onto torch's
# already-loaded runtime, so one process has exactly one HIP runtime.
# (The system ROCm userspace at /opt/rocm can fail to enumerate the GPU
# on some hosts while torch's bundled runtime works.)
try:
    import torch  # noqa: F401
except ImportError:
    pass

from gloo_amd._C import (  # noqa: F401
    Buffer,
    Context,
    Device,
    DType,
    EnforceNotMet,
    FileStore,
    GlooAmdError,
    HashStore,
    IoError,
    Pair,
    PrefixStore,
    ReduceOp,
    Store,
    TcpStore,
    TimeoutError,
    UnboundBuffer,
    allgather,
    allgatherv,
    allreduce,
    alltoall,
    alltoallv,
    barrier,
    broadcast,
    create_tcp_device,
    create_tls_device,
    dtype_size,
    gather,
    gatherv,
    reduce,
    reduce_scatter,
    scatter,
)

__version__ = "0.1.0"

_TORCH_DTYPE_MAP = None


def dtype_from_torch(dtype):
    """Map a torch.dtype to a gloo_amd DType."""
    global _TORCH_DTYPE_MAP
    if _TORCH_DTYPE_MAP is None:
        import torch

        _TORCH_DTYPE_MAP = {
            torch.float32: DType.f32,
            torch.float64: DType.f64,
            torch.float16: DType.f16,
            torch.bfloat16: DType.bf16,
            torch.int8: DType.i8,
            torch.uint8: DType.u8,
            torch.int32: DType.i32,
            torch.int64: DType.i64,
        }
    return _TORCH_DTYPE_MAP[dtype]


def dtype_from_numpy(dtype):
    """Map a numpy dtype to a gloo_amd DType."""
    import numpy as np

    return {
        np.dtype(np.float32): DType.f32,
        np.dtype(np.float64): DType.f64,
        np.dtype(np.float16): DType.f16,
        np.dtype(np.int8): DType.i8,
        np.dtype(np.uint8): DType.u8,
        np.dtype(np.int32): DType.i32,
        np.dtype(np.int64): DType.i64,
        np.dtype(np.uint64): DType.u64,
    }[np.dtype(dtype)]
