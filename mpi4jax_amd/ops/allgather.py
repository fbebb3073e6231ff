"""allgather.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/allgather.py``
(user fn :43-74; output shape ``(size, *x.shape)`` :124-128; no AD rules).
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ._common import prepare


def allgather(x, *, comm=None, token=NOTSET):
    """Gather ``x`` from every process onto every process.

    Arguments:
        x: tensor or scalar input; must have the same shape/dtype on all
           processes.
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor of shape ``(nproc, *x.shape)``.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "allgather")
    return backend.allgather(x.detach(), comm)
