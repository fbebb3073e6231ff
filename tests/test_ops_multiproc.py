"""Multi-process (gloo, world_size=2) semantics of all 12 ops.

The distributed path that runs on MI355X via RCCL shares all op-layer code
with this CPU path; only the transport differs, so these tests pin the
rank/shape/semantics contracts for both.
"""

import pytest
import torch

import mpi4jax_amd as m
from tests._mp import run_multiproc


def _all_collectives(rank, ws):
    x = torch.arange(6, dtype=torch.float32).reshape(2, 3) + rank
    ref = x.clone()

    # allreduce: sum over ranks
    y = m.allreduce(x, m.SUM)
    expect = sum(
        torch.arange(6, dtype=torch.float32).reshape(2, 3) + r
        for r in range(ws)
    )
    assert torch.equal(y, expect), (rank, "allreduce")

    assert torch.equal(m.allreduce(x, m.MAX),
                       torch.arange(6.).reshape(2, 3) + ws - 1)
    assert torch.allclose(m.allreduce(x, m.AVG), expect / ws)

    # allgather
    y = m.allgather(x)
    assert y.shape == (ws, 2, 3)
    for r in range(ws):
        assert torch.equal(y[r], torch.arange(6.).reshape(2, 3) + r)

    # alltoall: row i of rank r is r*10+i; after, row i of rank r is i*10+r
    a = torch.stack([torch.full((3,), float(rank * 10 + i))
                     for i in range(ws)])
    y = m.alltoall(a)
    for i in range(ws):
        assert torch.equal(y[i], torch.full((3,), float(i * 10 + rank)))

    # bcast
    y = m.bcast(x, 1)
    if rank == 1:
        assert y is x
    else:
        assert torch.equal(y, torch.arange(6.).reshape(2, 3) + 1)

    # gather
    y = m.gather(x, 0)
    if rank == 0:
        assert y.shape == (ws, 2, 3)
        for r in range(ws):
            assert torch.equal(y[r], torch.arange(6.).reshape(2, 3) + r)
    else:
        assert y is x  # non-root passthrough (gather.py:140-150)

    # scatter
    if rank == 0:
        src = torch.stack([torch.full((2, 3), float(r)) for r in range(ws)])
        y = m.scatter(src, 0)
    else:
        y = m.scatter(torch.empty(2, 3), 0)
    assert torch.equal(y, torch.full((2, 3), float(rank)))

    # reduce
    y = m.reduce(x, m.SUM, 0)
    if rank == 0:
        assert torch.equal(y, expect)
    else:
        assert y is x

    # scan (inclusive prefix)
    y = m.scan(x, m.SUM)
    expect_scan = sum(
        torch.arange(6.).reshape(2, 3) + r for r in range(rank + 1)
    )
    assert torch.equal(y, expect_scan)

    # send/recv pair
    status = m.Status()
    if rank == 0:
        m.send(x, 1, tag=3)
        got = m.recv(x, 1, tag=4, status=status)
        assert torch.equal(got, torch.arange(6.).reshape(2, 3) + 1)
        assert status.source == 1 and status.tag == 4
    else:
        got = m.recv(x, 0, tag=3)
        assert torch.equal(got, torch.arange(6.).reshape(2, 3))
        m.send(x, 0, tag=4)

    # sendrecv ring
    other = (rank + 1) % ws
    y = m.sendrecv(x, x, source=other, dest=other)
    assert torch.equal(y, torch.arange(6.).reshape(2, 3) + other)

    # input never mutated by any of the above
    assert torch.equal(x, ref)

    m.barrier()


def test_all_collectives():
    run_multiproc(_all_collectives, 2)


def _autograd(rank, ws):
    x = (torch.arange(4.0) + rank).requires_grad_()
    y = m.allreduce(x, m.SUM)
    g = torch.full((4,), float(rank + 1))
    y.backward(g)
    # VJP of allreduce(SUM) is the identity (per-rank cotangent unchanged)
    assert torch.equal(x.grad, g)

    # sendrecv grad routes the cotangent back along the reversed edge
    other = (rank + 1) % ws
    x2 = (torch.arange(3.0) + rank).requires_grad_()
    y2 = m.sendrecv(x2, x2.detach(), source=other, dest=other)
    assert torch.equal(y2, torch.arange(3.0) + other)
    y2.backward(torch.full((3,), float(rank)))
    # rank r receives the cotangent produced on rank `other`
    assert torch.equal(x2.grad, torch.full((3,), float(other)))


def test_autograd_multiproc():
    run_multiproc(_autograd, 2)


def _comm_management(rank, ws):
    world = m.get_world()
    assert world.Get_size() == ws and world.Get_rank() == rank

    clone = world.Clone()
    assert clone.Get_size() == ws
    x = torch.tensor([float(rank + 1)])
    assert m.allreduce(x, m.SUM, comm=clone).item() == sum(
        r + 1 for r in range(ws)
    )

    # Split into singleton comms by color=rank
    sub = world.Split(color=rank, key=0)
    assert sub.Get_size() == 1 and sub.Get_rank() == 0
    assert m.allreduce(x, m.SUM, comm=sub).item() == rank + 1

    # Split into one group
    sub2 = world.Split(color=0, key=-rank)  # reversed order
    assert sub2.Get_size() == ws
    assert sub2.Get_rank() == ws - 1 - rank

    # default comm is created lazily and cached
    d1 = m.get_default_comm()
    d2 = m.get_default_comm()
    assert d1 is d2


def test_comm_management():
    run_multiproc(_comm_management, 2)


def _barrier_ordering(rank, ws, path):
    import time

    if rank == 1:
        time.sleep(0.5)
    with open(f"{path}/r{rank}.start", "w") as f:
        f.write("s")
    m.barrier()
    # after the barrier every rank's start file must exist
    import os

    for r in range(ws):
        assert os.path.exists(f"{path}/r{r}.start"), (rank, r)


def test_barrier_ordering(tmp_path):
    run_multiproc(_barrier_ordering, 2, args=(str(tmp_path),))


def _world4(rank, ws):
    x = torch.tensor([1.0 + rank])
    assert m.allreduce(x, m.SUM).item() == sum(1.0 + r for r in range(ws))
    y = m.scan(x, m.SUM)
    assert y.item() == sum(1.0 + r for r in range(rank + 1))
    a = torch.arange(float(ws)) + rank * 100
    z = m.alltoall(a[:, None])
    for i in range(ws):
        assert z[i, 0].item() == rank + i * 100


@pytest.mark.slow
def test_world4():
    run_multiproc(_world4, 4)


def _reduce_scatter(rank, ws):
    src = torch.stack([torch.full((3,), float(r + 1)) for r in range(ws)])
    y = m.reduce_scatter(src + rank, m.SUM)
    # slice `rank` summed over ranks: ws*(rank+1) + sum(r)
    expect = ws * (rank + 1) + sum(range(ws))
    assert torch.equal(y, torch.full((3,), float(expect)))


def test_reduce_scatter_multiproc():
    run_multiproc(_reduce_scatter, 2)


def _nested_split(rank, ws):
    # Split of a Split must not hang (group creation is local-sync)
    world = m.get_world()
    sub = world.Split(color=0, key=rank)  # everyone
    sub2 = sub.Split(color=sub.rank % 2, key=0)  # split the sub-comm
    x = torch.tensor([1.0])
    total = m.allreduce(x, m.SUM, comm=sub2).item()
    assert total == sub2.size
    clone = sub2.Clone()
    assert m.allreduce(x, m.SUM, comm=clone).item() == sub2.size


def test_nested_split():
    run_multiproc(_nested_split, 2)


def _exotic_dtypes(rank, ws):
    # movement ops across the wire for dtypes MPI/gloo treat specially
    for dtype in (torch.bfloat16, torch.float16, torch.int16,
                  torch.complex64, torch.bool):
        if dtype == torch.bool:
            x = torch.tensor([rank % 2 == 0, True, False])
        elif dtype == torch.complex64:
            x = (torch.arange(6.) + 1j * rank).to(dtype)
        else:
            x = (torch.arange(6.) + rank).to(dtype)
        y = m.bcast(x.clone(), 0)
        g = m.allgather(x)
        assert g.dtype == dtype and g.shape[0] == ws
        a2a = m.alltoall(x[:ws].reshape(ws, 1).contiguous())
        assert a2a.dtype == dtype
        got = m.gather(x, 0)
        if rank == 0:
            assert got.shape[0] == ws
        if dtype not in (torch.bool, torch.complex64):
            s = m.scan(x, m.SUM)
            assert s.dtype == dtype


def test_exotic_dtypes_over_wire():
    run_multiproc(_exotic_dtypes, 2)


def _usage_global_sum(rank, ws):
    # the docs/usage.md headline example at world size 4
    a = torch.zeros(3, 3)
    result = m.allreduce(a + rank, m.SUM)
    assert torch.equal(result, torch.full((3, 3), float(sum(range(ws)))))


def test_usage_example_world4():
    run_multiproc(_usage_global_sum, 4)


def _bitwise(rank, ws):
    # per-rank distinct bit patterns
    x = torch.tensor([0b1100, 0b1010, 0b0110], dtype=torch.int32) << rank
    parts = [torch.tensor([0b1100, 0b1010, 0b0110], dtype=torch.int32) << r
             for r in range(ws)]

    def fold(op, seq):
        out = seq[0].clone()
        for t in seq[1:]:
            out = op(out, t)
        return out

    assert torch.equal(m.allreduce(x, m.BAND), fold(torch.bitwise_and,
                                                    parts))
    assert torch.equal(m.allreduce(x, m.BOR), fold(torch.bitwise_or,
                                                   parts))
    assert torch.equal(m.allreduce(x, m.BXOR), fold(torch.bitwise_xor,
                                                    parts))

    # inclusive prefix
    y = m.scan(x, m.BOR)
    assert torch.equal(y, fold(torch.bitwise_or, parts[:rank + 1])), rank

    # reduce to root 0
    y = m.reduce(x, m.BXOR, root=0)
    if rank == 0:
        assert torch.equal(y, fold(torch.bitwise_xor, parts))
    else:
        assert torch.equal(y, x)  # non-root passthrough

    # bool: bitwise == logical
    b = torch.tensor([rank == 0, True, False])
    assert torch.equal(m.allreduce(b, m.BAND),
                       torch.tensor([ws == 1, True, False]))
    assert torch.equal(m.allreduce(b, m.BOR),
                       torch.tensor([True, True, False]))

    # float dtype must be rejected
    with pytest.raises(ValueError):
        m.allreduce(torch.zeros(2), m.BAND)


def test_bitwise_reductions():
    run_multiproc(_bitwise, 2)


def test_bitwise_reductions_world4():
    run_multiproc(_bitwise, 4)


def _tagged_wildcards(rank, ws):
    """Tagged send + wildcard recv on the CPU backend (the gloo-tag
    implementation hung here and echoed the requested tag into Status —
    found by the master/worker example; fixed by riding the envelope
    plane).  Matching must be MPI's earliest-send rule with the ACTUAL
    tag in Status."""
    if rank == 0:
        m.send(torch.full((2,), 1.0), dest=1, tag=5)
        m.send(torch.full((3,), 2.0), dest=1, tag=9)
        m.send(torch.full((4,), 3.0), dest=1, tag=5)
    elif rank == 1:
        st = m.Status()
        y = m.recv(torch.empty(3), source=0, tag=9, status=st)
        assert torch.equal(y, torch.full((3,), 2.0))
        assert (st.source, st.tag) == (0, 9)
        st = m.Status()
        y = m.recv(torch.empty(2), source=m.ANY_SOURCE, tag=m.ANY_TAG,
                   status=st)
        assert torch.equal(y, torch.full((2,), 1.0)), "earliest send wins"
        assert (st.source, st.tag) == (0, 5)
        st = m.Status()
        y = m.recv(torch.empty(4), source=0, tag=m.ANY_TAG, status=st)
        assert torch.equal(y, torch.full((4,), 3.0))
        assert st.tag == 5
    # zero-size messages carry their envelope like any other
    if rank == 0:
        m.send(torch.empty(0), dest=1, tag=3)
    elif rank == 1:
        st = m.Status()
        y = m.recv(torch.empty(0), source=0, tag=m.ANY_TAG, status=st)
        assert y.numel() == 0 and st.tag == 3 and st.count == 0
    # sendrecv accepts ANY_SOURCE for its receive half (MPI_Sendrecv
    # semantics) on the CPU envelope plane
    peer = (rank + 1) % ws
    st = m.Status()
    y = m.sendrecv(torch.full((2,), float(rank)), torch.empty(2),
                   source=m.ANY_SOURCE, dest=peer, status=st)
    assert y[0].item() == float((rank - 1) % ws)
    assert st.source == (rank - 1) % ws
    # a wildcard SEND tag must raise on every backend
    import pytest as _pytest
    with _pytest.raises(ValueError, match="sendable"):
        m.send(torch.ones(1), dest=peer, tag=m.ANY_TAG)


def test_tagged_wildcard_matching_world2():
    run_multiproc(_tagged_wildcards, 2)


def _split_tagged(rank, ws):
    """Tagged wildcards on Split sub-communicators: each comm has its own
    envelope plane (per-comm gloo group + box), so identical tags on
    sibling comms never cross."""
    world = m.get_world()
    sub = world.Split(color=rank // 2)  # {0,1} and {2,3}
    peer = 1 - sub.rank
    if sub.rank == 0:
        m.send(torch.full((2,), float(rank)), dest=peer, tag=7, comm=sub)
        st = m.Status()
        y = m.recv(torch.empty(2), source=m.ANY_SOURCE, tag=m.ANY_TAG,
                   comm=sub, status=st)
    else:
        st = m.Status()
        y = m.recv(torch.empty(2), source=m.ANY_SOURCE, tag=m.ANY_TAG,
                   comm=sub, status=st)
        m.send(torch.full((2,), float(rank)), dest=peer, tag=7, comm=sub)
    exp = float((rank // 2) * 2 + peer)
    assert y[0].item() == exp, (rank, y[0].item(), exp)
    assert st.tag == 7 and st.source == peer


def test_split_comm_tagged_isolation_world4():
    run_multiproc(_split_tagged, 4)
