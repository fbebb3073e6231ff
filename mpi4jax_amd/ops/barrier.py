"""barrier.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/barrier.py``
(token-only primitive :42-56).  On the RCCL backend this enqueues a 4-byte
allreduce on the current HIP stream — a *stream* barrier across ranks: all
prior device work on every rank must complete before any rank's stream
proceeds past it.  For a host-visible barrier, follow with
``torch.cuda.synchronize()`` (the reference pairs it with
``jax.effects_barrier()`` the same way, test_barrier.py:38-50).
"""

from ..parallel.comm import resolve_comm
from ..utils.tokens import NOTSET, raise_if_token_is_set

import torch


def barrier(*, comm=None, token=NOTSET):
    """Synchronize all processes of ``comm``.

    Backend choice must be identical on every rank: the RCCL (stream)
    barrier is used only when this communicator already has an RCCL
    communicator — i.e. the program has been doing GPU collectives on it.
    Otherwise the bootstrap-plane (host) barrier runs; that also keeps
    CPU-only multi-process runs correct on machines that have a GPU.
    """
    raise_if_token_is_set(token)
    comm = resolve_comm(comm)
    if comm._rccl_id is not None and torch.cuda.is_initialized():
        from .._backend import rccl

        rccl.barrier(comm)
    else:
        from .._backend import cpu

        cpu.barrier(comm)
