"""scan — inclusive prefix reduction across ranks.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/scan.py``
(user fn :44-63; same output shape :113-114).  RCCL has no scan primitive;
the native path is a rank chain of ``ncclRecv`` → CDNA4 combine kernel →
``ncclSend``, all enqueued on the current HIP stream.
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from .reduce_ops import resolve_op
from ._common import prepare


def scan(x, op, *, comm=None, token=NOTSET):
    """Inclusive prefix reduction: rank r gets ``x_0 ⊕ ... ⊕ x_r``.

    Returns:
        Tensor: same shape as ``x``.
    """
    raise_if_token_is_set(token)
    op = resolve_op(op, "scan")
    from .reduce_ops import Op

    if op is Op.AVG:
        raise ValueError("scan: AVG is not a valid scan operator")
    x, comm, backend = prepare(x, comm, "scan")
    return backend.scan(x.detach(), op, comm)
