"""sendrecv — differentiable.

Reference: ``/root/reference/mpi4jax/_src/collective_ops/sendrecv.py``
(user fn :50-104; JVP binds a sendrecv on the tangents :237-275; the
transpose/VJP swaps source and dest :278-293).  Deadlock-free by
construction on the RCCL path: the send and recv are enqueued inside one
``ncclGroupStart``/``ncclGroupEnd``.
"""

import torch

from ..utils.tokens import NOTSET, raise_if_token_is_set
from ..utils.validation import enforce_types
from ..utils.status import ANY_SOURCE, ANY_TAG, Status
from ._common import prepare, as_tensor
from ..utils.dtypes import check_dtype


class _Sendrecv(torch.autograd.Function):
    @staticmethod
    def forward(sendbuf, recvbuf, source, dest, sendtag, recvtag, comm,
                status, backend):
        return backend.sendrecv(
            sendbuf, recvbuf, source, dest, sendtag, recvtag, comm, status
        )

    @staticmethod
    def setup_context(ctx, inputs, output):
        (sendbuf, recvbuf, source, dest, sendtag, recvtag, comm, _status,
         backend) = inputs
        ctx.meta = (source, dest, sendtag, recvtag, comm, backend)
        ctx.send_shape = tuple(sendbuf.shape)
        ctx.recv_shape = tuple(recvbuf.shape)

    @staticmethod
    def backward(ctx, grad):
        # VJP: route the output cotangent back along the reversed edge
        # (sendrecv.py:278-293 swaps source and dest).  Tags do not
        # identify cotangents — and the forward recvtag may be ANY_TAG,
        # which is not a sendable tag — so the backward pair always uses
        # (sendtag=0, recvtag=ANY_TAG): both sides agree by construction.
        source, dest, _sendtag, _recvtag, comm, backend = ctx.meta
        if source == ANY_SOURCE:
            raise RuntimeError(
                "grad through sendrecv(source=ANY_SOURCE) is undefined — "
                "the reversed edge is unknown; pass an explicit source"
            )
        template = grad.new_empty(ctx.send_shape)
        grad_send = backend.sendrecv(
            grad.contiguous(), template, dest, source, 0, ANY_TAG,
            comm, None,
        )
        return grad_send, None, None, None, None, None, None, None, None

    @staticmethod
    def jvp(ctx, send_t, recv_t, *_):
        # tangent flows along the same edge (sendrecv.py:237-275)
        source, dest, sendtag, recvtag, comm, backend = ctx.meta
        if send_t is None:
            raise RuntimeError(
                "sendrecv jvp requires a tangent for sendbuf"
            )
        # the tangent arrives shaped like the OUTPUT (= recvbuf)
        template = send_t.new_empty(ctx.recv_shape)
        return backend.sendrecv(
            send_t.contiguous(), template, source, dest, sendtag, recvtag,
            comm, None,
        )


@enforce_types(source=int, dest=int, sendtag=int, recvtag=int,
               status=(type(None), Status))
def sendrecv(sendbuf, recvbuf, source, dest, *, sendtag=0, recvtag=ANY_TAG,
             comm=None, status=None, token=NOTSET):
    """Send ``sendbuf`` to ``dest`` while receiving from ``source``.

    Arguments:
        sendbuf: tensor or scalar to send.
        recvbuf: template with the shape/dtype to receive (not modified).
        source: rank to receive from.
        dest: rank to send to.
        sendtag / recvtag: message tags.
        comm: the communicator (defaults to a clone of the world).
        status: optional :class:`Status`.

    Returns:
        Tensor: the received data (a new tensor).
    """
    raise_if_token_is_set(token)
    sendbuf, comm, backend = prepare(sendbuf, comm, "sendrecv")
    recvbuf = as_tensor(recvbuf, "sendrecv")
    check_dtype(recvbuf, "sendrecv")
    if recvbuf.device != sendbuf.device:
        raise ValueError(
            f"sendrecv: sendbuf ({sendbuf.device}) and recvbuf template "
            f"({recvbuf.device}) must live on the same device"
        )
    for name, r in (("source", source), ("dest", dest)):
        if name == "source" and r == ANY_SOURCE:
            continue  # MPI_Sendrecv accepts ANY_SOURCE for the receive
        if not 0 <= r < comm.size:
            raise ValueError(
                f"sendrecv: invalid {name} {r} for comm size {comm.size}"
            )
    return _Sendrecv.apply(
        sendbuf, recvbuf.detach(), source, dest, sendtag, recvtag, comm,
        status, backend,
    )
