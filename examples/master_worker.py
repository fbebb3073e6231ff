"""Master/worker task farm over ANY_SOURCE + tags — the classic MPI
dynamic-load-balancing idiom (reference-era mpi4jax exposed it through
``recv(source=-1)`` on the CPU path; ``mpi_ops_common.h:354-367``
transmits source and tag through MPI).

Rank 0 hands out work items; workers return results tagged with the item
id; the master receives with ``ANY_SOURCE``/``ANY_TAG`` and learns who
finished what from the synthesized ``Status`` — so fast workers pull more
work.  On CPU (gloo) this runs as-is; on the RCCL path set
``MPI4JAX_AMD_GPU_ENVELOPE=1`` (docs/sharp-bits.md) to enable the same
wildcard matching over the gloo envelope plane.

    python -m mpi4jax_amd.run -n 4 examples/master_worker.py --items 24
"""

import argparse

import torch

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import mpi4jax_amd as m  # noqa: E402

TAG_WORK = 1
TAG_STOP = 2


def master(comm, n_items, size):
    torch.manual_seed(0)
    items = [torch.randn(64) for _ in range(n_items)]
    results = {}
    next_item = 0
    # prime every worker with one item
    for w in range(1, size):
        if next_item < n_items:
            m.send(items[next_item], dest=w, tag=next_item + 10, comm=comm)
            next_item += 1
        else:
            m.send(torch.zeros(64), dest=w, tag=TAG_STOP, comm=comm)
    live = min(size - 1, n_items)
    while live:
        st = m.Status()
        out = m.recv(torch.empty(1), source=m.ANY_SOURCE, tag=m.ANY_TAG,
                     comm=comm, status=st)
        results[st.tag - 10] = out.item()  # tag carries the item id
        if next_item < n_items:  # the finisher pulls the next item
            m.send(items[next_item], dest=st.source,
                   tag=next_item + 10, comm=comm)
            next_item += 1
        else:
            m.send(torch.zeros(64), dest=st.source, tag=TAG_STOP,
                   comm=comm)
            live -= 1
    # verify against the local computation
    for k, v in results.items():
        exp = (items[k] ** 2).sum().item()
        assert abs(v - exp) < 1e-3 * (1 + abs(exp)), (k, v, exp)
    print(f"master: {len(results)}/{n_items} items verified OK",
          flush=True)


def worker(comm):
    done = 0
    while True:
        st = m.Status()
        x = m.recv(torch.empty(64), source=0, tag=m.ANY_TAG, comm=comm,
                   status=st)
        if st.tag == TAG_STOP:
            break
        m.send((x ** 2).sum().reshape(1), dest=0, tag=st.tag, comm=comm)
        done += 1
    print(f"worker {comm.rank}: {done} items", flush=True)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--items", type=int, default=24)
    args = p.parse_args()
    m.init()
    comm = m.get_world()
    if comm.size < 2:
        print("needs >= 2 ranks (run under mpi4jax_amd.run -n 2)")
        return
    if comm.rank == 0:
        master(comm, args.items, comm.size)
    else:
        worker(comm)
    m.finalize()


if __name__ == "__main__":
    main()
