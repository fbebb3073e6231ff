"""gather.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/gather.py``
(user fn :44-89; root receives ``(nproc, *shape)``, non-root ranks get
their input back unchanged :140-150).  RCCL lacks a gather primitive; the
native path is grouped ``ncclSend`` to root + root-side ``ncclRecv`` ×size.
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ._common import prepare


@enforce_types(root=int)
def gather(x, root, *, comm=None, token=NOTSET):
    """Gather ``x`` from all processes onto ``root``.

    ``x`` must have the same shape and dtype on all processes.

    Returns:
        Tensor: on root, shape ``(nproc, *x.shape)``; elsewhere ``x``
        unchanged.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "gather")
    res = backend.gather(x.detach(), root, comm)
    if comm.rank != root:
        return x
    return res
