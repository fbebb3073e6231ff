"""Sequence-parallel (Ulysses-style) shard swaps on alltoall.

The reference's distributed-transpose pattern (reshape → alltoall →
reshape, ``tests/collective_ops/test_alltoall.py:43-65``) is the building
block of Ulysses sequence parallelism: attention wants heads sharded and
the full sequence local; the rest of the model wants the sequence sharded.
These helpers swap between the two layouts over the xGMI clique (grouped
per-peer p2p — the bandwidth-optimal alltoall on a fully-connected
topology).
"""

from .comm import resolve_comm


def seq_to_head_shard(x, comm=None):
    """(seq/P, H, D) per rank → (seq, H/P, D) per rank."""
    from ..ops.alltoall import alltoall  # circular-import guard

    comm = resolve_comm(comm)
    p = comm.size
    s_loc, h, d = x.shape
    assert h % p == 0, f"heads {h} not divisible by {p} ranks"
    # split heads into P destination chunks
    chunks = x.reshape(s_loc, p, h // p, d).permute(1, 0, 2, 3).contiguous()
    out = alltoall(chunks, comm=comm)  # (P, seq/P, H/P, D)
    return out.reshape(p * s_loc, h // p, d)


def head_to_seq_shard(x, comm=None):
    """(seq, H/P, D) per rank → (seq/P, H, D) per rank."""
    from ..ops.alltoall import alltoall  # circular-import guard

    comm = resolve_comm(comm)
    p = comm.size
    s, h_loc, d = x.shape
    assert s % p == 0, f"sequence {s} not divisible by {p} ranks"
    chunks = x.reshape(p, s // p, h_loc, d).contiguous()
    out = alltoall(chunks, comm=comm)  # (P, seq/P, H/P, D)
    return (out.permute(1, 0, 2, 3)
            .reshape(s // p, p * h_loc, d).contiguous())
