"""reduce_scatter — MI355X-native extra (no reference equivalent).

The reference exposes only the 12 MPI ops; ``MPI_Reduce_scatter`` is not
among them.  RCCL has a native ``ncclReduceScatter`` that is the building
block of bucketed data-parallel training, so it is exposed here as an
extension: input ``(nproc, ...)``, rank r receives the reduction of slice
r across all ranks.
"""

import torch

from ..utils.tokens import NOTSET, raise_if_token_is_set
from .reduce_ops import resolve_op
from ._common import prepare


def reduce_scatter(x, op, *, comm=None, token=NOTSET):
    """Reduce ``x`` across processes and scatter the leading axis.

    Arguments:
        x: tensor of shape ``(nproc, ...)`` (same on all processes).
        op: reduction operator.
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor of shape ``x.shape[1:]``: the reduction of slice ``rank``.
    """
    raise_if_token_is_set(token)
    op = resolve_op(op, "reduce_scatter")
    x, comm, backend = prepare(x, comm, "reduce_scatter")
    if x.ndim == 0 or x.shape[0] != comm.size:
        raise ValueError(
            f"reduce_scatter input must have shape (nproc, ...), got "
            f"{tuple(x.shape)} with nproc={comm.size}"
        )
    x = x.detach()
    if x.is_cuda:
        from .._backend import rccl
        from ..ops.reduce_ops import RCCL_OP_ENUM, BITWISE_OP_ENUM

        rccl._check_op(op, x, "reduce_scatter")
        if op in BITWISE_OP_ENUM:
            # no RCCL bitwise: full allreduce composition, then slice
            full = rccl.allreduce(x, op, comm)
            return full[comm.rank].clone()
        # _reduction_view maps bool onto uint8 MAX/MIN (logical OR/AND),
        # upcasts int16 (no RCCL int16), views complex as real pairs —
        # same dtype shims as allreduce/reduce/scan
        xr, rop, post = rccl._reduction_view(x.contiguous(), op,
                                             "reduce_scatter")
        out = torch.empty(tuple(xr.shape[1:]), dtype=xr.dtype,
                          device=xr.device)
        rccl.ext().reduce_scatter(out.reshape(-1), xr.reshape(-1),
                                  RCCL_OP_ENUM[rop], comm.rccl_handle())
        if post is not None:
            out = post(out)
        if x.is_complex():
            out = torch.view_as_complex(out)
        return out.reshape(tuple(x.shape[1:]))
    # CPU: allreduce then take this rank's slice (gloo has no native
    # reduce_scatter for all our dtypes)
    from .._backend import cpu

    full = cpu.allreduce(x, op, comm)
    return full[comm.rank].clone()
