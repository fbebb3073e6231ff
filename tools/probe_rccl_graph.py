"""Probe: do RCCL enqueues capture into a hipGraph and replay correctly?

World-1 check of the machinery behind multi-rank graph capture
(make_stepper): an ncclAllReduce, a grouped enqueue, and a staging kernel
are captured into one torch.cuda.CUDAGraph and replayed; replay must
recompute from the live buffer contents.

Run: python tools/probe_rccl_graph.py   (single GPU)
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

import mpi4jax_amd as m  # noqa: E402
from mpi4jax_amd._backend import rccl  # noqa: E402
from mpi4jax_amd.parallel.comm import get_default_comm  # noqa: E402


def main():
    m.init()
    x = torch.ones(1 << 20, device="cuda")
    y = m.allreduce(x, m.SUM)  # init RCCL communicator eagerly
    torch.cuda.synchronize()
    print("eager allreduce ok:", bool((y == 1).all()))

    ext = rccl.ext()
    comm_id = get_default_comm().rccl_handle()
    out = torch.empty_like(x)

    graph = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(graph):
            ext.allreduce(out, x, 0, comm_id)  # SUM
            ext.allreduce(out, out, 0, comm_id)
    except Exception as e:
        print("CAPTURE FAILED:", repr(e))
        return 1
    print("capture ok")

    x.fill_(3.0)
    graph.replay()
    torch.cuda.synchronize()
    ok1 = bool((out == 3.0).all())
    x.fill_(-2.5)
    graph.replay()
    torch.cuda.synchronize()
    ok2 = bool((out == -2.5).all())
    print("replay recomputes from live buffers:", ok1 and ok2,
          float(out[0]))

    # grouped enqueue + barrier inside a second graph
    g2 = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g2):
            ext.group_start()
            ext.allreduce(out, x, 2, comm_id)  # MIN
            ext.group_end()
            ext.barrier(comm_id)
    except Exception as e:
        print("GROUPED CAPTURE FAILED:", repr(e))
        return 1
    x.fill_(7.0)
    g2.replay()
    torch.cuda.synchronize()
    print("grouped replay ok:", bool((out == 7.0).all()))
    print("PROBE PASS")
    return 0


if __name__ == "__main__":
    sys.exit(main())
