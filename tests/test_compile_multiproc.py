"""Compiled collectives at world_size > 1 (gloo).

Round-1 compile tests ran only at world 1; these pin the
``torch.compile(fullgraph=True)`` custom-op path — values, autograd, and
cross-rank matching — at world 2 over the CPU transport (same op-layer
code as the RCCL path).  Reference analog: the whole suite under
``mpirun -np 2`` (docs/developers.rst:18-27).
"""

import torch

from tests._mp import run_multiproc


def _compiled_allreduce_worker(rank, ws):
    from mpi4jax_amd import jit_ops

    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, "sum") * 2.0

    x = torch.full((8,), float(rank + 1))
    y = f(x)
    expect = 2.0 * sum(range(1, ws + 1))
    assert torch.equal(y, torch.full((8,), expect)), y[0]

    # autograd through the compiled collective
    @torch.compile(fullgraph=True)
    def g(x):
        return jit_ops.allreduce(x, "sum").sum()

    w = torch.randn(5, requires_grad=True)
    g(w).backward()
    assert torch.equal(w.grad, torch.ones(5))  # identity VJP


def test_compiled_allreduce_world2():
    run_multiproc(_compiled_allreduce_worker, 2)


def _compiled_sendrecv_worker(rank, ws):
    from mpi4jax_amd import jit_ops

    src = (rank - 1) % ws
    dst = (rank + 1) % ws

    @torch.compile(fullgraph=True)
    def ring(x):
        return jit_ops.sendrecv(x, x, src, dst)

    x = torch.full((4,), float(rank), requires_grad=True)
    y = ring(x)
    assert y[0].item() == src, y

    # VJP routes the cotangent along the reversed edge: each rank's grad
    # is the cotangent its *dest* saw (all-ones here)
    y.sum().backward()
    assert torch.equal(x.grad, torch.ones(4))


def test_compiled_sendrecv_ring_world2():
    run_multiproc(_compiled_sendrecv_worker, 2)


def _compiled_gather_scatter_worker(rank, ws):
    from mpi4jax_amd import jit_ops

    @torch.compile(fullgraph=True)
    def f(x, s):
        g = jit_ops.gather(x, 0)
        a = jit_ops.allgather(x)
        t = jit_ops.alltoall(s)
        b = jit_ops.bcast(x, 1)
        return g, a, t, b

    x = torch.full((3,), float(rank))
    s = torch.arange(float(ws)).reshape(ws, 1) + rank * ws
    g, a, t, b = f(x, s)
    if rank == 0:
        for k in range(ws):
            assert g[k, 0].item() == k
    else:
        assert torch.equal(g, torch.zeros(ws, 3))  # zero-filled non-root
    for k in range(ws):
        assert a[k, 0].item() == k
        assert t[k, 0].item() == k * ws + rank
    assert b[0].item() == 1.0


def test_compiled_gather_scatter_world2():
    run_multiproc(_compiled_gather_scatter_worker, 2)


def _compiled_scan_reduce_worker(rank, ws):
    from mpi4jax_amd import jit_ops

    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.scan(x, "sum"), jit_ops.reduce(x, "sum", 0)

    x = torch.full((6,), float(rank + 1))
    sc, rd = f(x)
    assert sc[0].item() == sum(range(1, rank + 2))
    if rank == 0:
        assert rd[0].item() == sum(range(1, ws + 1))


def test_compiled_scan_reduce_world2():
    run_multiproc(_compiled_scan_reduce_worker, 2)


def _vmap_worker(rank, ws):
    from mpi4jax_amd import jit_ops

    # fast registered rule: elementwise across the batch
    x = torch.full((3, 5), float(rank + 1))
    y = torch.func.vmap(lambda v: jit_ops.allreduce(v, "sum"))(x)
    assert torch.equal(y, torch.full((3, 5), float(sum(range(1, ws + 1)))))

    # grad through vmap'd allreduce
    w = torch.randn(3, 5, requires_grad=True)
    torch.func.vmap(lambda v: jit_ops.allreduce(v, "sum"))(w).sum().backward()
    assert torch.equal(w.grad, torch.ones(3, 5))

    # ops without a rule take torch's per-slice loop fallback — the loop
    # count is the (identical) batch size on every rank, so messages still
    # match; values must be the stacked per-slice results
    g = torch.func.vmap(lambda v: jit_ops.allgather(v))(x)
    assert g.shape == (3, ws, 5)
    for k in range(ws):
        assert g[0, k, 0].item() == k + 1


def test_vmap_collectives_world2():
    run_multiproc(_vmap_worker, 2)
