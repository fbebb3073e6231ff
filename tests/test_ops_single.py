"""Single-process semantics of all 12 ops (a 'cluster of one').

Mirrors the reference's single-process leg of its dual-mode suite
(SURVEY.md §4): collectives must self-test degenerate cases at size 1.
"""

import pytest
import torch

import mpi4jax_amd as m


@pytest.fixture
def x():
    return torch.arange(12, dtype=torch.float32).reshape(3, 4)


def test_allreduce(x):
    y = m.allreduce(x, m.SUM)
    assert torch.equal(y, x)
    assert y is not x  # immutability: new tensor


def test_allreduce_ops(x):
    for op in (m.PROD, m.MIN, m.MAX, m.AVG):
        assert torch.equal(m.allreduce(x, op), x)


def test_allreduce_scalar():
    y = m.allreduce(3.0, m.SUM)
    assert y.item() == 3.0


def test_allgather(x):
    y = m.allgather(x)
    assert y.shape == (1, 3, 4)
    assert torch.equal(y[0], x)


def test_alltoall():
    x = torch.arange(4.0).reshape(1, 4)
    y = m.alltoall(x)
    assert torch.equal(y, x)
    with pytest.raises(ValueError):
        m.alltoall(torch.zeros(2, 3))  # wrong leading axis


def test_barrier():
    m.barrier()


def test_bcast(x):
    y = m.bcast(x, 0)
    assert y is x  # root gets its input back (reference bcast.py:124-133)


def test_gather(x):
    y = m.gather(x, 0)
    assert y.shape == (1, 3, 4)
    assert torch.equal(y[0], x)


def test_scatter(x):
    y = m.scatter(x[None], 0)
    assert torch.equal(y, x)
    with pytest.raises(ValueError):
        m.scatter(torch.zeros(2, 3), 0)


def test_reduce(x):
    y = m.reduce(x, m.SUM, 0)
    assert torch.equal(y, x)


def test_scan(x):
    y = m.scan(x, m.SUM)
    assert torch.equal(y, x)


def test_sendrecv_self(x):
    y = m.sendrecv(x, x, source=0, dest=0)
    assert torch.equal(y, x)
    assert y is not x


def test_send_recv_self(x):
    m.send(x, 0, tag=7)
    status = m.Status()
    y = m.recv(x, 0, tag=7, status=status)
    assert torch.equal(y, x)
    assert status.source == 0
    assert status.tag == 7
    assert status.count == x.numel() * x.element_size()


def test_input_never_mutated(x):
    ref = x.clone()
    m.allreduce(x, m.SUM)
    m.allgather(x)
    m.bcast(x, 0)
    m.gather(x, 0)
    m.reduce(x, m.SUM, 0)
    m.scan(x, m.SUM)
    assert torch.equal(x, ref)


def test_token_rejected(x):
    with pytest.raises(RuntimeError, match="token"):
        m.allreduce(x, m.SUM, token=object())


def test_bad_rank(x):
    with pytest.raises(ValueError):
        m.send(x, 5)
    with pytest.raises(ValueError):
        m.recv(x, 5)
    with pytest.raises(ValueError):
        m.sendrecv(x, x, source=0, dest=3)


def test_bad_op(x):
    with pytest.raises(TypeError):
        m.allreduce(x, "not-a-reduction")
    with pytest.raises(TypeError):
        m.allreduce(x, 123)


def test_capability_probes():
    assert isinstance(m.has_rccl_support(), bool)
    assert m.has_cuda_support() == m.has_rccl_support()


def test_reduce_scatter(x):
    y = m.reduce_scatter(x[None], m.SUM)
    assert torch.equal(y, x)
    with pytest.raises(ValueError):
        m.reduce_scatter(torch.zeros(3, 2), m.SUM)


def test_batched_collective_idiom():
    """The reference registers vmap batching rules; here a batched
    collective IS the collective on the stacked tensor (PARITY.md)."""
    xs = torch.randn(4, 3, 2)  # a "batch" of 4 inputs
    stacked = m.allreduce(xs, m.SUM)
    looped = torch.stack([m.allreduce(x, m.SUM) for x in xs])
    assert torch.equal(stacked, looped)


def test_noncontiguous_inputs():
    a = torch.arange(12.0).reshape(3, 4)
    t = a.t()  # non-contiguous view
    y = m.allreduce(t, m.SUM)
    assert y.shape == (4, 3) and torch.equal(y, t)
    g = m.allgather(t)
    assert g.shape == (1, 4, 3) and torch.equal(g[0], t)
    s = m.sendrecv(t, t, source=0, dest=0)
    assert torch.equal(s, t)
