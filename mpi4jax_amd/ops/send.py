"""send.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/send.py``
(user fn :44-68; token-only output :118-124).  The RCCL path enqueues a
grouped ``ncclSend`` on the current HIP stream; tags are accepted for API
parity but not transmitted (RCCL has no envelope — ordering is stream
order, which is exactly what the reference's token enforced).
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ._common import prepare


@enforce_types(dest=int, tag=int)
def send(x, dest, *, tag=0, comm=None, token=NOTSET):
    """Send ``x`` to rank ``dest``.

    Arguments:
        x: tensor or scalar to send (never mutated).
        dest: destination rank.
        tag: message tag.
        comm: the communicator (defaults to a clone of the world).
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "send")
    if not 0 <= dest < comm.size:
        raise ValueError(f"send: invalid dest {dest} for comm size {comm.size}")
    backend.send(x.detach(), dest, tag, comm)
