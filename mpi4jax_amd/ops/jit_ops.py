"""torch.compile-traceable collectives.

The reference's headline capability is communication *inside* ``jax.jit``.
This framework's equivalents are (a) hipGraph capture of eager regions
(see ``ShallowWater.make_stepper``) and (b) these ``torch.library`` custom
ops, which trace into ``torch.compile`` graphs with ``fullgraph=True`` —
no graph breaks — including autograd (allreduce-SUM identity VJP and the
sendrecv source/dest swap, same rules as the eager ops).

Usage::

    from mpi4jax_amd import jit_ops

    @torch.compile(fullgraph=True)
    def step(x):
        return jit_ops.allreduce(x, "sum").sum()

Communicators cross the graph boundary as int registry keys (the same
int64-handle marshalling idea the reference uses for MPI objects,
``mpi_ops_common.h:36-48``): pass ``comm=None`` for the default, or
``comm=jit_ops.comm_key(my_comm)`` computed outside the compiled region —
the key is runtime data, so one graph serves any communicator.

Not traced: ``send``/``recv`` (their buffered self-messaging queue is
process state a traced graph must not capture) and ``barrier`` (returns
nothing; call it outside the compiled region).  Use :func:`sendrecv` for
paired transfers inside graphs.
"""

import torch

from ..parallel.comm import Communicator, resolve_comm
from .reduce_ops import Op
from .._backend import backend_for

# int key <-> Communicator registry (keys are stable for a process life)
_COMMS = {}
_KEYS = {}
_NEXT_KEY = [1]


def comm_key(comm) -> int:
    """Registry key for a communicator.

    Inside a ``torch.compile(fullgraph=True)`` region pass either
    ``comm=None`` (the default communicator, resolved once at trace time)
    or this function's result computed OUTSIDE the graph — int keys flow
    through the graph as runtime data, so one compiled function serves any
    number of communicators without recompilation.
    """
    comm = resolve_comm(comm)
    if id(comm) not in _KEYS:
        key = _NEXT_KEY[0]
        _NEXT_KEY[0] += 1
        _COMMS[key] = comm
        _KEYS[id(comm)] = key
    return _KEYS[id(comm)]


@torch._dynamo.assume_constant_result
def _default_key() -> int:
    # zero-arg: safe for assume_constant_result; runs eagerly at trace
    # time (communicator creation is untraceable), baked per-graph
    return comm_key(None)


def _key_of(comm) -> int:
    if comm is None:
        return _default_key()
    if isinstance(comm, int):
        return comm
    return comm_key(comm)


def _comm(key: int) -> Communicator:
    c = _COMMS.get(key)
    if c is None:
        # finalize() cleared the registry but a cached compiled graph still
        # holds this baked key: remap it to the (re-initialized) default
        # communicator so old graphs keep working after finalize()+init().
        c = resolve_comm(None)
        _COMMS[key] = c
        _KEYS[id(c)] = key
    return c


_OPS = {o.value: o for o in Op}


# ----------------------------------------------------------------- allreduce
@torch.library.custom_op("mpi4jax_amd::allreduce", mutates_args=())
def _allreduce(x: torch.Tensor, op: str, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).allreduce(x.contiguous(), _OPS[op], comm)


@_allreduce.register_fake
def _(x, op, key):
    return torch.empty_like(x)


def _allreduce_setup(ctx, inputs, output):
    ctx.op = inputs[1]


def _allreduce_bwd(ctx, grad):
    if ctx.op != "sum":
        raise RuntimeError("allreduce is only differentiable for op=sum")
    return grad, None, None  # identity VJP (reference allreduce.py:152-159)


torch.library.register_autograd("mpi4jax_amd::allreduce", _allreduce_bwd,
                                setup_context=_allreduce_setup)


def _allreduce_vmap(info, in_dims, x, op, key):
    # the reference registers a batching rule for allreduce
    # (allreduce.py:132-135): the op is elementwise across the batch, so
    # reducing the whole batched tensor is the batched op; the batch dim
    # stays where it was.  Other collectives reshape across ranks, where
    # batching changes the wire layout — vmap over them raises torch's
    # standard "no vmap rule" error.
    return _allreduce(x.contiguous(), op, key), in_dims[0]


torch.library.register_vmap("mpi4jax_amd::allreduce", _allreduce_vmap)


def allreduce(x, op=Op.SUM, *, comm=None):
    op = op.value if isinstance(op, Op) else str(op)
    return _allreduce(x, op, _key_of(comm))


# ----------------------------------------------------------------- allgather
@torch.library.custom_op("mpi4jax_amd::allgather", mutates_args=())
def _allgather(x: torch.Tensor, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).allgather(x.contiguous(), comm)


@_allgather.register_fake
def _(x, key):
    size = _comm(key).size
    return x.new_empty((size,) + tuple(x.shape))


def allgather(x, *, comm=None):
    return _allgather(x, _key_of(comm))


# ----------------------------------------------------------------- alltoall
@torch.library.custom_op("mpi4jax_amd::alltoall", mutates_args=())
def _alltoall(x: torch.Tensor, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).alltoall(x.contiguous(), comm)


@_alltoall.register_fake
def _(x, key):
    return torch.empty_like(x)


def alltoall(x, *, comm=None):
    return _alltoall(x, _key_of(comm))


# ------------------------------------------------------------ reduce_scatter
@torch.library.custom_op("mpi4jax_amd::reduce_scatter", mutates_args=())
def _reduce_scatter(x: torch.Tensor, op: str, key: int) -> torch.Tensor:
    from .reduce_scatter import reduce_scatter as rs

    return rs(x, _OPS[op], comm=_comm(key))


@_reduce_scatter.register_fake
def _(x, op, key):
    return x.new_empty(tuple(x.shape[1:]))


def reduce_scatter(x, op=Op.SUM, *, comm=None):
    op = op.value if isinstance(op, Op) else str(op)
    return _reduce_scatter(x, op, _key_of(comm))


# ----------------------------------------------------------------- bcast
@torch.library.custom_op("mpi4jax_amd::bcast", mutates_args=())
def _bcast(x: torch.Tensor, root: int, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).bcast(x.contiguous(), root, comm)


@_bcast.register_fake
def _(x, root, key):
    return torch.empty_like(x)


def bcast(x, root, *, comm=None):
    return _bcast(x, root, _key_of(comm))


# ----------------------------------------------------------------- scan
@torch.library.custom_op("mpi4jax_amd::scan", mutates_args=())
def _scan(x: torch.Tensor, op: str, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).scan(x.contiguous(), _OPS[op], comm)


@_scan.register_fake
def _(x, op, key):
    return torch.empty_like(x)


def scan(x, op=Op.SUM, *, comm=None):
    op = op.value if isinstance(op, Op) else str(op)
    return _scan(x, op, _key_of(comm))


# ----------------------------------------------------------------- sendrecv
@torch.library.custom_op("mpi4jax_amd::sendrecv", mutates_args=())
def _sendrecv(sendbuf: torch.Tensor, recvbuf: torch.Tensor, source: int,
              dest: int, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(sendbuf).sendrecv(
        sendbuf.contiguous(), recvbuf, source, dest, 0, -1, comm, None
    )


@_sendrecv.register_fake
def _(sendbuf, recvbuf, source, dest, key):
    return torch.empty_like(recvbuf)


def _sendrecv_setup(ctx, inputs, output):
    _, recvbuf, source, dest, key = inputs
    ctx.meta = (source, dest, key)
    ctx.send_shape = tuple(inputs[0].shape)


def _sendrecv_bwd(ctx, grad):
    source, dest, key = ctx.meta
    template = grad.new_empty(ctx.send_shape)
    # VJP routes the cotangent along the reversed edge
    # (reference sendrecv.py:278-293)
    g = _sendrecv(grad.contiguous(), template, dest, source, key)
    return g, None, None, None, None


torch.library.register_autograd("mpi4jax_amd::sendrecv", _sendrecv_bwd,
                                setup_context=_sendrecv_setup)


def sendrecv(sendbuf, recvbuf, source, dest, *, comm=None):
    return _sendrecv(sendbuf, recvbuf, source, dest, _key_of(comm))


# ----------------------------------------------------------------- reduce
@torch.library.custom_op("mpi4jax_amd::reduce", mutates_args=())
def _reduce(x: torch.Tensor, op: str, root: int, key: int) -> torch.Tensor:
    comm = _comm(key)
    out = backend_for(x).reduce(x.contiguous(), _OPS[op], root, comm)
    return out if out is not None else x.clone()


@_reduce.register_fake
def _(x, op, root, key):
    return torch.empty_like(x)


def reduce(x, op, root, *, comm=None):
    op = op.value if isinstance(op, Op) else str(op)
    return _reduce(x, op, root, _key_of(comm))


# ----------------------------------------------------------------- gather
@torch.library.custom_op("mpi4jax_amd::gather", mutates_args=())
def _gather(x: torch.Tensor, root: int, key: int) -> torch.Tensor:
    comm = _comm(key)
    out = backend_for(x).gather(x.contiguous(), root, comm)
    if out is None:  # non-root: keep a static output shape for the graph
        # zero-filled, not uninitialized — reading it is still meaningless
        # on non-root, but the value is at least deterministic
        out = x.new_zeros((comm.size,) + tuple(x.shape))
    return out


@_gather.register_fake
def _(x, root, key):
    return x.new_empty((_comm(key).size,) + tuple(x.shape))


def gather(x, root, *, comm=None):
    """Root gets ``(nproc, *shape)``; other ranks get a zero-filled tensor
    of that shape (static shapes are required inside a compiled graph —
    the eager op's non-root input passthrough does not translate)."""
    return _gather(x, root, _key_of(comm))


# ----------------------------------------------------------------- scatter
@torch.library.custom_op("mpi4jax_amd::scatter", mutates_args=())
def _scatter(x: torch.Tensor, root: int, key: int) -> torch.Tensor:
    comm = _comm(key)
    return backend_for(x).scatter(x.contiguous(), root, comm)


@_scatter.register_fake
def _(x, root, key):
    comm = _comm(key)
    shape = tuple(x.shape[1:]) if comm.rank == root else tuple(x.shape)
    return x.new_empty(shape)


def scatter(x, root, *, comm=None):
    return _scatter(x, root, _key_of(comm))
