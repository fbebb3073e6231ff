"""Shared helpers for the op layer."""

import torch

from ..parallel.comm import resolve_comm
from ..utils.dtypes import check_dtype
from .._backend import backend_for


def as_tensor(x, op_name):
    """Accept tensors and Python/numpy scalars (the reference ops accept
    scalars too — e.g. test_allreduce.py scalar cases)."""
    if isinstance(x, torch.Tensor):
        return x
    try:
        return torch.as_tensor(x)
    except Exception:
        raise TypeError(
            f"{op_name}: expected a torch.Tensor or scalar, got {type(x)}"
        )


def prepare(x, comm, op_name):
    x = as_tensor(x, op_name)
    check_dtype(x, op_name)
    comm = resolve_comm(comm)
    return x, comm, backend_for(x)
