"""allreduce — differentiable for op=SUM.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/allreduce.py``
(user fn :45-70; JVP :138-149 reduces the tangent; transpose :152-159 is the
identity).  Autodiff here is a ``torch.autograd.Function`` with the same
rules: backward (VJP/transpose) is the identity, forward-mode ``jvp``
allreduces the tangent.
"""

import torch

from .reduce_ops import Op, resolve_op
from ..utils.tokens import NOTSET, raise_if_token_is_set
from ._common import prepare


class _AllreduceSum(torch.autograd.Function):
    @staticmethod
    def forward(x, comm, backend):
        return backend.allreduce(x, Op.SUM, comm)

    @staticmethod
    def setup_context(ctx, inputs, output):
        _, comm, backend = inputs
        ctx.comm = comm
        ctx.backend = backend

    @staticmethod
    def backward(ctx, grad):
        # transpose of allreduce(SUM) is the identity — the cotangent is
        # already replicated across ranks (allreduce.py:152-159).
        return grad, None, None

    @staticmethod
    def jvp(ctx, x_t, _c, _b):
        # tangent of allreduce(SUM) is allreduce(tangent) (:138-149)
        return ctx.backend.allreduce(x_t, Op.SUM, ctx.comm)


class _NonDifferentiable(torch.autograd.Function):
    @staticmethod
    def forward(x, op, comm, backend):
        return backend.allreduce(x, op, comm)

    @staticmethod
    def setup_context(ctx, inputs, output):
        ctx.op = inputs[1]

    @staticmethod
    def backward(ctx, grad):
        raise RuntimeError(
            f"allreduce is only differentiable for op=SUM, got {ctx.op}"
        )


def allreduce(x, op, *, comm=None, token=NOTSET):
    """Perform an allreduce operation.

    Differentiable (backward and forward mode) when ``op`` is ``SUM``.

    Arguments:
        x: tensor or scalar input (never mutated).
        op: reduction operator (``mpi4jax_amd.SUM`` etc.).
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor: result of the reduction, same shape as ``x``.
    """
    raise_if_token_is_set(token)
    op = resolve_op(op, "allreduce")
    x, comm, backend = prepare(x, comm, "allreduce")
    if op is Op.SUM:
        return _AllreduceSum.apply(x, comm, backend)
    if torch.is_grad_enabled() and x.requires_grad:
        return _NonDifferentiable.apply(x, op, comm, backend)
    return backend.allreduce(x, op, comm)
