"""alltoall.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/alltoall.py``
(user fn :43-74; requires ``x.shape[0] == nproc`` :65-67; per-process count
is ``prod(shape[1:])`` :87; no AD rules).  On the RCCL backend this lowers
to grouped ``ncclSend``/``ncclRecv`` over the xGMI clique (RCCL has no
alltoall primitive) — one direct point-to-point message per peer, which is
exactly the right shape for the fully-connected 8-GPU topology.
"""

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ._common import prepare


def alltoall(x, *, comm=None, token=NOTSET):
    """Exchange the i-th slice of ``x`` with process i.

    Arguments:
        x: tensor whose leading axis equals the number of processes.
        comm: the communicator (defaults to a clone of the world).

    Returns:
        Tensor of the same shape; slice ``i`` holds data from process i.
    """
    raise_if_token_is_set(token)
    x, comm, backend = prepare(x, comm, "alltoall")
    if x.ndim == 0 or x.shape[0] != comm.size:
        raise ValueError(
            f"alltoall input must have shape (nproc, ...), got "
            f"{tuple(x.shape)} with nproc={comm.size}"
        )
    return backend.alltoall(x.detach(), comm)
