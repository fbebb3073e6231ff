"""Dtype support tables.

The reference supports 14 numpy dtypes via ``MPI_TYPE_MAP``
(``/root/reference/mpi4jax/_src/utils.py:101-116``) but has no bf16/f16
(MPI lacks them).  RCCL supports them natively, and bf16 is in the
benchmark configs, so this table extends the reference's coverage.

Complex dtypes have no RCCL type; for SUM and all data-movement ops they are
handled by viewing the buffer as (2x) real elements — valid because complex
addition is elementwise real addition.  PROD/MIN/MAX on complex are rejected
(MIN/MAX are not even well-defined; the reference inherits whatever MPI does,
we make it explicit).
"""

import torch

# dtypes that can travel on both backends
SUPPORTED_DTYPES = (
    torch.float32,
    torch.float64,
    torch.float16,
    torch.bfloat16,
    torch.int8,
    torch.uint8,
    torch.int16,
    torch.int32,
    torch.int64,
    torch.bool,
    torch.complex64,
    torch.complex128,
)

COMPLEX_DTYPES = (torch.complex64, torch.complex128)

# view used for reductions/transport of complex data
COMPLEX_AS_REAL = {
    torch.complex64: torch.float32,
    torch.complex128: torch.float64,
}

# nccl/rccl enum values (rccl.h ncclDataType_t) — used by the native ext path.
# Kept in Python so the op layer can validate before crossing into C++.
RCCL_DTYPE_ENUM = {
    torch.int8: 0,  # ncclInt8
    torch.uint8: 1,  # ncclUint8
    torch.int32: 2,  # ncclInt32
    torch.int64: 4,  # ncclInt64
    torch.float16: 6,  # ncclFloat16
    torch.float32: 7,  # ncclFloat32
    torch.float64: 8,  # ncclFloat64
    torch.bfloat16: 9,  # ncclBfloat16
}


def check_dtype(x: torch.Tensor, op_name: str):
    if x.dtype not in SUPPORTED_DTYPES:
        raise TypeError(
            f"{op_name}: unsupported dtype {x.dtype}. Supported: "
            f"{[str(d) for d in SUPPORTED_DTYPES]}"
        )
