"""Math-integration tests: distributed linear algebra on the collectives.

Mirrors the reference's matvec/CG suites
(``tests/collective_ops/test_allreduce_matvec.py:41-72`` — column-partitioned
matvec whose transpose tower alternates allreduce and slicing — and
``tests/test_jax_transforms.py:6-22`` — a matrix-free distributed CG solve).
"""

import torch

import mpi4jax_amd as m
from tests._mp import run_multiproc


def _partitioned_matvec(rank, ws):
    torch.manual_seed(0)
    n = 8
    A = torch.randn(n, n, dtype=torch.float64)
    x = torch.randn(n, dtype=torch.float64)

    # column-partition A over ranks; each rank holds A[:, cols] and the
    # matching slice of x; full result needs an allreduce(SUM)
    cols = slice(rank * n // ws, (rank + 1) * n // ws)
    y_local = A[:, cols] @ x[cols]
    y = m.allreduce(y_local, m.SUM)
    assert torch.allclose(y, A @ x, atol=1e-12)

    # row-partition: matvec then allgather
    rows = slice(rank * n // ws, (rank + 1) * n // ws)
    y_rows = m.allgather(A[rows] @ x).reshape(-1)
    assert torch.allclose(y_rows, A @ x, atol=1e-12)


def test_partitioned_matvec():
    run_multiproc(_partitioned_matvec, 2)


def _distributed_cg(rank, ws):
    """Matrix-free CG on an SPD system with distributed dot products."""
    torch.manual_seed(1)
    n = 16
    Q = torch.linalg.qr(torch.randn(n, n, dtype=torch.float64))[0]
    A = Q @ torch.diag(torch.linspace(1, 10, n, dtype=torch.float64)) @ Q.T
    b = torch.randn(n, dtype=torch.float64)

    # each rank owns a block of rows; global dots via allreduce
    rows = slice(rank * n // ws, (rank + 1) * n // ws)

    def dist_dot(u_loc, v_loc):
        return m.allreduce(u_loc @ v_loc, m.SUM)

    def dist_matvec(x_full):
        # local rows of A @ x, then allgather to the full vector
        return m.allgather(A[rows] @ x_full).reshape(-1)

    x = torch.zeros(n, dtype=torch.float64)
    r = b.clone()
    p = r.clone()
    rs = dist_dot(r[rows], r[rows])
    for _ in range(2 * n):
        Ap = dist_matvec(p)
        alpha = rs / dist_dot(p[rows], Ap[rows])
        x = x + alpha * p
        r = r - alpha * Ap
        rs_new = dist_dot(r[rows], r[rows])
        if rs_new.item() < 1e-24:
            break
        p = r + (rs_new / rs) * p
        rs = rs_new
    assert torch.allclose(A @ x, b, atol=1e-8), (A @ x - b).abs().max()


def test_distributed_cg():
    run_multiproc(_distributed_cg, 2)


def _grad_through_matvec(rank, ws):
    """Autograd through communication: d/dx of sum(allreduce(A_loc x_loc))."""
    torch.manual_seed(2)
    n = 6
    A = torch.randn(n, n, dtype=torch.float64)
    cols = slice(rank * n // ws, (rank + 1) * n // ws)
    x_loc = torch.randn(n // ws, dtype=torch.float64, requires_grad=True)
    y = m.allreduce(A[:, cols] @ x_loc, m.SUM)
    y.sum().backward()
    # identity VJP: cotangent of y is ones on every rank
    expect = A[:, cols].T @ torch.ones(n, dtype=torch.float64)
    assert torch.allclose(x_loc.grad, expect, atol=1e-12)


def test_grad_through_matvec():
    run_multiproc(_grad_through_matvec, 2)


def test_distributed_transpose_pattern():
    """reshape → alltoall → reshape distributed transpose (the reference's
    SP/Ulysses building block, test_alltoall.py:43-65) — single rank."""
    x = torch.arange(12.0).reshape(1, 3, 4)
    y = m.alltoall(x)
    assert torch.equal(y, x)


def _distributed_transpose(rank, ws):
    """Row-partitioned global matrix transposed via alltoall."""
    p, q = 3, 4 * ws
    torch.manual_seed(7)
    M = torch.randn(ws * p, q)
    local = M[rank * p:(rank + 1) * p]  # (p, q)

    # split columns into per-destination chunks, exchange, reassemble
    chunks = local.reshape(p, ws, q // ws).permute(1, 0, 2).contiguous()
    got = m.alltoall(chunks)  # (ws, p, q//ws): block r of rank r's rows
    mine = got.reshape(ws * p, q // ws).t().contiguous()  # (q//ws, ws*p)

    expect = M.t()[rank * (q // ws):(rank + 1) * (q // ws)]
    assert torch.equal(mine, expect), (rank, (mine - expect).abs().max())


def test_distributed_transpose_multiproc():
    run_multiproc(_distributed_transpose, 2)


def _ddp_average(rank, ws):
    """Bucketed gradient averaging == manual average across ranks."""
    torch.manual_seed(10 + rank)
    from mpi4jax_amd.parallel import average_gradients

    model = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.Tanh(), torch.nn.Linear(16, 2)
    ).double()
    # identical weights everywhere (seeded per-rank inputs, shared init)
    for p in model.parameters():
        p.data.copy_(m.bcast(p.data, 0))
    x = torch.randn(4, 8, dtype=torch.float64)
    model(x).sum().backward()
    grads_local = [p.grad.clone() for p in model.parameters()]
    average_gradients(model.parameters())
    for g_loc, p in zip(grads_local, model.parameters()):
        expect = m.allreduce(g_loc, m.SUM) / ws
        assert torch.allclose(p.grad, expect, atol=1e-12)


def test_ddp_average_gradients():
    run_multiproc(_ddp_average, 2)
