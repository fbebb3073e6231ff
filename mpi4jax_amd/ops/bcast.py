"""bcast.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/bcast.py``
(user fn :44-75; on the root the wrapper returns ``x`` itself, non-root
ranks get the received array :124-133).
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ._common import prepare


@enforce_types(root=int)
def bcast(x, root, *, comm=None, token=NOTSET):
    """Broadcast ``x`` from ``root`` to all processes.

    Arguments:
        x: tensor or scalar; on non-root ranks only shape/dtype are used.
        root: rank that provides the data.
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor: on the root, ``x`` unchanged; elsewhere the received data.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "bcast")
    res = backend.bcast(x.detach(), root, comm)
    if comm.rank == root:
        return x
    return res
