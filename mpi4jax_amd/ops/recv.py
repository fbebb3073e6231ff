"""recv.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/recv.py``
(user fn :47-84; ``x`` is a shape/dtype template only and is never
overwritten; optional Status out-param :100-103).  Status fields are
synthesized (RCCL has no envelope — SURVEY.md §2.3 #11): shapes are static
so source/count are always known; ANY_SOURCE is CPU-backend only.
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ..utils.status import ANY_SOURCE, ANY_TAG, Status
from ._common import prepare


@enforce_types(source=int, tag=int, status=(type(None), Status))
def recv(x, source=ANY_SOURCE, *, tag=ANY_TAG, comm=None, status=None,
         token=NOTSET):
    """Receive a message into a *new* tensor shaped like ``x``.

    Arguments:
        x: template tensor with the shape/dtype to receive (not modified).
        source: source rank (ANY_SOURCE only on the CPU backend).
        tag: message tag.
        comm: the communicator (defaults to a clone of the world).
        status: optional :class:`Status` filled with source/tag/count.

    Returns:
        Tensor: the received data.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "recv")
    if source != ANY_SOURCE and not 0 <= source < comm.size:
        raise ValueError(
            f"recv: invalid source {source} for comm size {comm.size}"
        )
    return backend.recv(x.detach(), source, tag, comm, status)
