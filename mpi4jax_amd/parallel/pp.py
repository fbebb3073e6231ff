"""Pipeline-parallel activation transfer with gradient routing.

Completes the parallelism-pattern suite (DP: ``ddp``, TP: ``tp``, SP:
``sp``): point-to-point activation sends whose backward passes route the
gradient back along the reversed edge — the same transpose convention as
:func:`mpi4jax_amd.sendrecv` (reference sendrecv.py:278-293), applied to
one-directional stage boundaries.

Usage (two stages)::

    # stage 0                          # stage 1
    h = layer0(x)                      a = recv_activation(tmpl, src=0)
    out = send_activation(h, dest=1)   loss = layer1(a).sum()
    backward_send(out)                 loss.backward()   # sends grad(a) back

``backward_send`` runs the sending stage's backward once the downstream
stage computes gradients (its seed is ignored — the true output gradient
arrives from ``dest``).  Scheduling (micro-batches, 1F1B) is left to the
caller; these primitives make the transfers autograd-correct.
"""

import torch

from .comm import resolve_comm


class _SendActivation(torch.autograd.Function):
    @staticmethod
    def forward(x, dest, comm):
        from .._backend import backend_for

        backend_for(x).send(x.detach().contiguous(), dest, 0, comm)
        return x

    @staticmethod
    def setup_context(ctx, inputs, output):
        _, dest, comm = inputs
        ctx.dest = dest
        ctx.comm = comm

    @staticmethod
    def backward(ctx, seed):
        # the real gradient comes from the downstream stage; the local
        # seed (whatever backward_send passed) is discarded
        from .._backend import backend_for

        g = backend_for(seed).recv(seed, ctx.dest, 0, ctx.comm, None)
        return g, None, None


class _RecvActivation(torch.autograd.Function):
    # `anchor` is a zero-size leaf requiring grad: the received tensor has
    # no local autograd ancestry, so the anchor keeps the edge alive and
    # backward reaches this Function (where the gradient is sent upstream).
    @staticmethod
    def forward(template, anchor, source, comm):
        from .._backend import backend_for

        return backend_for(template).recv(template.detach(), source, 0,
                                          comm, None)

    @staticmethod
    def setup_context(ctx, inputs, output):
        _, _, source, comm = inputs
        ctx.source = source
        ctx.comm = comm

    @staticmethod
    def backward(ctx, grad):
        from .._backend import backend_for

        backend_for(grad).send(grad.contiguous(), ctx.source, 0, ctx.comm)
        return None, None, None, None


def send_activation(x, dest, comm=None):
    """Send ``x`` to the next stage; returns ``x`` (identity) so the
    sending stage can later run :func:`backward_send` on it."""
    return _SendActivation.apply(x, dest, resolve_comm(comm))


def recv_activation(template, source, comm=None):
    """Receive an activation shaped like ``template``; differentiable —
    its backward sends the gradient back to ``source``."""
    anchor = torch.zeros(0, dtype=template.dtype, requires_grad=True)
    return _RecvActivation.apply(template, anchor, source,
                                 resolve_comm(comm))


def backward_send(sent):
    """Run the sending stage's backward; the output gradient is received
    from the downstream stage (the local seed is ignored)."""
    sent.backward(torch.empty_like(sent))
