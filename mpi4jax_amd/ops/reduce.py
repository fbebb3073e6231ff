"""reduce.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/reduce.py``
(user fn :45-73; non-root ranks get their input back unchanged :124-133).
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from .reduce_ops import resolve_op
from ._common import prepare


@enforce_types(root=int)
def reduce(x, op, root, *, comm=None, token=NOTSET):
    """Reduce ``x`` across processes onto ``root``.

    Returns:
        Tensor: on root, the reduction result; elsewhere ``x`` unchanged.
    """
    raise_if_token_is_set(token)
    op = resolve_op(op, "reduce")
    x, comm, backend = prepare(x, comm, "reduce")
    res = backend.reduce(x.detach(), op, root, comm)
    if comm.rank != root:
        return x
    return res
