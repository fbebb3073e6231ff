"""Matrix-free distributed conjugate-gradient solve.

The reference demonstrates this pattern in its test suite
(``/root/reference/tests/test_jax_transforms.py:6-22``): a CG iteration
whose matvec and dot products are distributed, with ``allreduce`` as the
only communication.  Here each rank owns a block of rows; global dot
products are ``allreduce(SUM)`` and the matvec gathers the full iterate
with ``allgather`` — both enqueued on the GPU stream when run on MI355X.

    python -m mpi4jax_amd.run -n 4 examples/distributed_cg.py --n 4096
"""

import argparse
import time

import torch

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import mpi4jax_amd as m


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=1024)
    p.add_argument("--tol", type=float, default=1e-10)
    args = p.parse_args()

    m.init()
    comm = m.get_world()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    n, ws, rank = args.n, comm.size, comm.rank
    assert n % ws == 0
    rows = slice(rank * n // ws, (rank + 1) * n // ws)

    # SPD system (same seed everywhere -> consistent global operator)
    torch.manual_seed(0)
    A_full = torch.randn(n, n, dtype=torch.float64)
    A = (A_full @ A_full.T / n + torch.eye(n, dtype=torch.float64))
    A_local = A[rows].to(device)
    b = torch.randn(n, dtype=torch.float64, device=device)

    def matvec(x):
        return m.allgather(A_local @ x, comm=comm).reshape(-1)

    def dot(u, v):
        return m.allreduce(u[rows] @ v[rows], m.SUM, comm=comm)

    x = torch.zeros_like(b)
    r = b.clone()
    pvec = r.clone()
    rs = dot(r, r)
    t0 = time.perf_counter()
    it = 0
    for it in range(10 * n):
        Ap = matvec(pvec)
        alpha = rs / dot(pvec, Ap)
        x = x + alpha * pvec
        r = r - alpha * Ap
        rs_new = dot(r, r)
        if rs_new.item() < args.tol:
            break
        pvec = r + (rs_new / rs) * pvec
        rs = rs_new
    if device == "cuda":
        torch.cuda.synchronize()
    wall = time.perf_counter() - t0

    resid = (matvec(x) - b).norm().item()
    if rank == 0:
        print(f"n={n} ranks={ws} device={device}: {it + 1} iterations, "
              f"residual {resid:.2e}, {wall:.3f}s")


if __name__ == "__main__":
    main()
