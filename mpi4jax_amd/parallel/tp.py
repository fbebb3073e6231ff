"""Tensor-parallel building blocks on the collectives.

SURVEY.md §2.4: the reference must be "a first-class communication backend
that DP/TP/SP frameworks could be built on".  This module is the TP proof:
the Megatron-style f/g conjugate pair and column/row-parallel linear
layers, with gradients flowing through mpi4jax_amd ops.

* ``g`` (reduce forward, identity backward) is exactly
  :func:`mpi4jax_amd.allreduce` with ``SUM`` — the reference's autodiff
  convention (identity transpose) is precisely the TP "g" operator.
* ``f`` (identity forward, reduce backward) is :func:`copy_to_parallel`.
"""

import torch

from .comm import resolve_comm


class _CopyToParallel(torch.autograd.Function):
    """Identity forward; allreduce(SUM) backward (Megatron's ``f``)."""

    @staticmethod
    def forward(x, comm):
        return x

    @staticmethod
    def setup_context(ctx, inputs, output):
        ctx.comm = inputs[1]

    @staticmethod
    def backward(ctx, grad):
        from ..ops.reduce_ops import Op
        from .._backend import backend_for

        return backend_for(grad).allreduce(grad.contiguous(), Op.SUM,
                                           ctx.comm), None


def copy_to_parallel(x, comm=None):
    """Enter a tensor-parallel region: identity now, gradient allreduce."""
    return _CopyToParallel.apply(x, resolve_comm(comm))


class ColumnParallelLinear(torch.nn.Module):
    """Linear layer with the output dimension sharded across ranks.

    Holds ``out_features / nproc`` rows of the weight.  With
    ``gather_output=True`` the full output is allgathered (inference
    convenience; gradients do not flow through the gather).
    """

    def __init__(self, in_features, out_features, *, comm=None, bias=True,
                 gather_output=False, dtype=torch.float32):
        super().__init__()
        self.comm = resolve_comm(comm)
        assert out_features % self.comm.size == 0
        self.out_local = out_features // self.comm.size
        self.linear = torch.nn.Linear(in_features, self.out_local,
                                      bias=bias, dtype=dtype)
        self.gather_output = gather_output

    def forward(self, x):
        y = self.linear(copy_to_parallel(x, self.comm))
        if self.gather_output:
            from ..ops.allgather import allgather

            parts = allgather(y.detach(), comm=self.comm)
            y = parts.movedim(0, -2).reshape(*y.shape[:-1], -1)
        return y


class RowParallelLinear(torch.nn.Module):
    """Linear layer with the input dimension sharded across ranks.

    Each rank computes a partial product from its input shard; the
    partials are summed with a (differentiable) allreduce.
    """

    def __init__(self, in_features, out_features, *, comm=None, bias=True,
                 dtype=torch.float32):
        super().__init__()
        self.comm = resolve_comm(comm)
        assert in_features % self.comm.size == 0
        self.in_local = in_features // self.comm.size
        self.linear = torch.nn.Linear(self.in_local, out_features,
                                      bias=False, dtype=dtype)
        self.bias = (torch.nn.Parameter(torch.zeros(out_features,
                                                    dtype=dtype))
                     if bias else None)

    def forward(self, x_shard):
        from ..ops.allreduce import allreduce
        from ..ops.reduce_ops import Op

        y = allreduce(self.linear(x_shard), Op.SUM, comm=self.comm)
        if self.bias is not None:
            y = y + self.bias
        return y
