"""Multi-process test helper.

Mirrors the reference's strategy of running the same assertions at any
nproc: worker functions receive (rank, world_size) and run
the full check; the helper spawns them over gloo on 127.0.0.1.
"""

import os
import socket

import torch.multiprocessing as mp


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _bootstrap(rank, world_size, port, fn, args):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import mpi4jax_amd as m

    m.init()
    try:
        fn(rank, world_size, *args)
    finally:
        m.finalize()


def run_multiproc(fn, world_size=2, args=(), timeout=240):
    """Spawn `world_size` processes running fn(rank, world_size, *args).

    Enforces a wall-clock timeout so a deadlocked worker fails the test
    instead of hanging the whole suite.
    """
    import time

    port = _free_port()
    ctx = mp.start_processes(
        _bootstrap,
        args=(world_size, port, fn, args),
        nprocs=world_size,
        join=False,
        start_method="spawn",
    )
    deadline = time.time() + timeout
    while not ctx.join(timeout=5):
        if time.time() > deadline:
            for p in ctx.processes:
                if p.is_alive():
                    p.terminate()
            raise TimeoutError(
                f"multiprocess test exceeded {timeout}s (deadlock?)"
            )
