"""Data-parallel gradient averaging built on the collectives.

The reference positions itself as the communication layer that
data/model-parallel frameworks are built on; this module
is that pattern made concrete for torch training loops: bucketed
``allreduce(AVG)`` over parameter gradients, enqueued on the compute
stream (zero-sync on GPU).

Usage::

    loss.backward()
    mpi4jax_amd.parallel.average_gradients(model.parameters())
    optimizer.step()
"""

import torch

from .comm import resolve_comm


def average_gradients(parameters, *, comm=None, bucket_cap_mb=64):
    """Allreduce-average ``.grad`` across ranks, bucketed by dtype/device.

    Buckets keep collective count low (per-link-bound xGMI favors fewer,
    larger messages — docs/tuning.md).  No-op at world size 1.
    """
    comm = resolve_comm(comm)
    if comm.size == 1:
        return
    params = [p for p in parameters if p.grad is not None]
    if not params:
        return
    cap = int(bucket_cap_mb * 1024 * 1024)

    buckets = {}
    for p in params:
        buckets.setdefault((p.grad.dtype, p.grad.device), []).append(p)

    for (_, _), ps in buckets.items():
        group, size = [], 0
        for p in ps + [None]:
            if p is not None:
                group.append(p)
                size += p.grad.numel() * p.grad.element_size()
            if p is None or size >= cap:
                if group:
                    _allreduce_bucket(group, comm)
                group, size = [], 0


def _allreduce_bucket(params, comm):
    # imported here to avoid a circular import at package init
    from ..ops.allreduce import allreduce
    from ..ops.reduce_ops import Op

    grads = [p.grad for p in params]
    flat = torch.cat([g.reshape(-1) for g in grads])
    flat = allreduce(flat.detach(), Op.AVG, comm=comm)
    offset = 0
    for p, g in zip(params, grads):
        n = g.numel()
        p.grad = flat[offset:offset + n].view_as(g)
        offset += n
