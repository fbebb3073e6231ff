"""Collective micro-benchmarks matching BASELINE.json's configs.

* ``allreduce``  — in-stream allreduce bandwidth sweep, headline point
  256 MiB bf16 (config #2: "allreduce 256 MiB bf16 inside jax.jit on
  2×MI355X" — here: enqueued on the torch compute stream via RCCL).
* ``bisection`` — alltoall + allgather at 1 GiB/rank (config #4, xGMI
  bisection bandwidth on 8 GPUs).
* ``grad``      — backward through allreduce(SUM) bf16 (config #5).

Run (N ranks, one per GPU):

    python -m mpi4jax_amd.run -n 8 benchmarks/bench_collectives.py [which]

Each rank-0 line is one JSON record.  busbw follows the standard
nccl-tests convention (allreduce: 2(n-1)/n, allgather/alltoall: (n-1)/n).
"""

import argparse
import json
import time

import torch

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import mpi4jax_amd as m


def _timeit(fn, steps, warmup, comm):
    for _ in range(warmup):
        fn()
    m.barrier(comm=comm)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    m.barrier(comm=comm)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    dt_t = torch.tensor([dt], dtype=torch.float64)
    return m.allreduce(dt_t, m.MAX, comm=comm).item()


def bench_allreduce(comm, device, steps=20, warmup=5):
    n = comm.size
    out = []
    for mib in (1, 16, 64, 256, 1024):
        nbytes = mib * 1024 * 1024
        x = torch.randn(nbytes // 2, device=device).to(torch.bfloat16)

        def fn():
            m.allreduce(x, m.SUM, comm=comm)

        dt = _timeit(fn, steps, warmup, comm)
        algbw = nbytes / dt / 1e9
        busbw = algbw * (2 * (n - 1) / n) if n > 1 else algbw
        rec = {"bench": "allreduce", "bytes": nbytes, "dtype": "bf16",
               "n_gpus": n, "time_us": round(dt * 1e6, 1),
               "algbw_GBps": round(algbw, 2), "busbw_GBps": round(busbw, 2)}
        out.append(rec)
        if comm.rank == 0:
            print(json.dumps(rec), flush=True)
    return out


def bench_bisection(comm, device, steps=10, warmup=3, gib_per_rank=1):
    n = comm.size
    nbytes = int(gib_per_rank * 1024 * 1024 * 1024)
    elems = nbytes // 2
    x = torch.randn(elems, device=device).to(torch.bfloat16)
    xa = x.reshape(n, -1)

    recs = []
    for name, fn, factor in (
        ("alltoall", lambda: m.alltoall(xa, comm=comm), (n - 1) / n),
        ("allgather", lambda: m.allgather(x[: elems // max(n, 1)], comm=comm),
         (n - 1) / n),
    ):
        dt = _timeit(fn, steps, warmup, comm)
        algbw = nbytes / dt / 1e9
        busbw = algbw * factor if n > 1 else algbw
        rec = {"bench": name, "bytes_per_rank": nbytes, "dtype": "bf16",
               "n_gpus": n, "time_us": round(dt * 1e6, 1),
               "algbw_GBps": round(algbw, 2), "busbw_GBps": round(busbw, 2)}
        recs.append(rec)
        if comm.rank == 0:
            print(json.dumps(rec), flush=True)
    return recs


def bench_reduce_scatter(comm, device, steps=20, warmup=5, mib=256):
    """reduce_scatter bandwidth (the DP gradient-sharding primitive)."""
    n = comm.size
    nbytes = mib * 1024 * 1024
    elems = nbytes // 2
    x = torch.randn(elems, device=device).to(torch.bfloat16).reshape(n, -1)

    def fn():
        m.reduce_scatter(x, m.SUM, comm=comm)

    dt = _timeit(fn, steps, warmup, comm)
    algbw = nbytes / dt / 1e9
    busbw = algbw * ((n - 1) / n) if n > 1 else algbw
    rec = {"bench": "reduce_scatter", "bytes": nbytes, "dtype": "bf16",
           "n_gpus": n, "time_us": round(dt * 1e6, 1),
           "algbw_GBps": round(algbw, 2), "busbw_GBps": round(busbw, 2)}
    if comm.rank == 0:
        print(json.dumps(rec), flush=True)
    return rec


def bench_grad(comm, device, steps=20, warmup=5, mib=256):
    """Backward through allreduce(SUM): fwd allreduce + identity VJP."""
    nbytes = mib * 1024 * 1024
    x = torch.randn(nbytes // 2, device=device).to(torch.bfloat16)
    x.requires_grad_()
    g = torch.ones_like(x)

    def fn():
        y = m.allreduce(x, m.SUM, comm=comm)
        y.backward(g)
        x.grad = None

    dt = _timeit(fn, steps, warmup, comm)
    rec = {"bench": "grad_allreduce", "bytes": nbytes, "dtype": "bf16",
           "n_gpus": comm.size, "time_us": round(dt * 1e6, 1),
           "fwd_bwd_GBps": round(nbytes / dt / 1e9, 2)}
    if comm.rank == 0:
        print(json.dumps(rec), flush=True)
    return rec


def main():
    p = argparse.ArgumentParser()
    p.add_argument("which", nargs="?", default="all",
                   choices=["all", "allreduce", "bisection", "grad",
                            "reduce_scatter"])
    p.add_argument("--gib", type=float, default=1.0)
    args = p.parse_args()
    m.init()
    comm = m.get_world()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if args.which in ("all", "allreduce"):
        bench_allreduce(comm, device)
    if args.which in ("all", "bisection"):
        bench_bisection(comm, device, gib_per_rank=args.gib)
    if args.which in ("all", "reduce_scatter"):
        bench_reduce_scatter(comm, device)
    if args.which in ("all", "grad"):
        bench_grad(comm, device)


if __name__ == "__main__":
    main()
