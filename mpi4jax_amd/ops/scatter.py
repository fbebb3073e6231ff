"""scatter.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/scatter.py``
(user fn :44-92; root input must be ``(nproc, ...)`` :80-84; output shape
is ``x.shape[1:]`` on root and ``x.shape`` on other ranks :145-153 — the
non-root input is a shape/dtype template).
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ._common import prepare


@enforce_types(root=int)
def scatter(x, root, *, comm=None, token=NOTSET):
    """Scatter the leading axis of root's ``x`` across processes.

    Arguments:
        x: on root, a tensor of shape ``(nproc, ...)``; on other ranks a
           template with the shape/dtype of one slice.
        root: rank that provides the data.
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor: this process's slice.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "scatter")
    if comm.rank == root:
        if x.ndim == 0 or x.shape[0] != comm.size:
            raise ValueError(
                f"scatter input on root must have shape (nproc, ...), got "
                f"{tuple(x.shape)} with nproc={comm.size}"
            )
    return backend.scatter(x.detach(), root, comm)
