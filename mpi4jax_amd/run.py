"""Process launcher: ``python -m mpi4jax_amd.run -n N script.py [args...]``.

The mpirun-equivalent for this framework (the reference is launched with
``mpirun -n N``): one process per GPU on a single node, rendezvous over
127.0.0.1, delegating to ``torch.distributed.run`` (RANK / WORLD_SIZE /
LOCAL_RANK / MASTER_* appear in each process's env; ``mpi4jax_amd.init()``
reads them).
"""

import sys


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    nproc = 1
    if argv and argv[0] in ("-n", "-np", "--nproc"):
        if len(argv) < 2:
            print("usage: python -m mpi4jax_amd.run -n N script.py [args...]",
                  file=sys.stderr)
            return 2
        nproc = int(argv[1])
        argv = argv[2:]
    if not argv:
        print("usage: python -m mpi4jax_amd.run -n N script.py [args...]",
              file=sys.stderr)
        return 2

    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]

    # child processes get sys.path[0] = the USER script's directory, so a
    # source-tree (non-installed) mpi4jax_amd would not be importable for
    # scripts outside this tree — propagate our package location
    import os

    pkg_parent = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    existing = os.environ.get("PYTHONPATH", "")
    if pkg_parent not in existing.split(os.pathsep):
        os.environ["PYTHONPATH"] = (
            pkg_parent + (os.pathsep + existing if existing else "")
        )

    from torch.distributed.run import main as torchrun

    torchrun([
        "--nnodes=1",
        f"--nproc-per-node={nproc}",
        "--master-addr=127.0.0.1",
        f"--master-port={port}",
        "--node-rank=0",
        *argv,
    ])
    return 0


if __name__ == "__main__":
    sys.exit(main())
