"""Property-based op tests (hypothesis): single-process semantics hold for
arbitrary shapes/dtypes — allreduce(SUM) is the identity at size 1, data
movement round-trips, scan is the identity, immutability always holds."""

import hypothesis
from hypothesis import given, settings, strategies as st
import torch

import mpi4jax_amd as m

DTYPES = [torch.float32, torch.float64, torch.float16, torch.bfloat16,
          torch.int8, torch.int32, torch.int64, torch.uint8]

shapes = st.lists(st.integers(1, 9), min_size=0, max_size=4).map(tuple)


def make(shape, dtype, seed):
    g = torch.Generator().manual_seed(seed)
    if dtype.is_floating_point:
        return torch.randn(shape, generator=g).to(dtype)
    return torch.randint(-9, 9, shape, generator=g).to(dtype)


@settings(max_examples=60, deadline=None)
@given(shape=shapes, dtype=st.sampled_from(DTYPES),
       seed=st.integers(0, 2**31 - 1))
def test_size1_semantics(shape, dtype, seed):
    x = make(shape, dtype, seed)
    ref = x.clone()

    assert torch.equal(m.allreduce(x, m.SUM), x)
    assert torch.equal(m.scan(x, m.SUM), x)
    ag = m.allgather(x)
    assert ag.shape == (1,) + shape and torch.equal(ag[0], x)
    assert torch.equal(m.scatter(m.gather(x, 0), 0), x)
    assert m.bcast(x, 0) is x
    got = m.sendrecv(x, x, source=0, dest=0)
    assert torch.equal(got, x) and got is not x
    # immutability after everything
    assert torch.equal(x, ref)


@settings(max_examples=40, deadline=None)
@given(shape=st.lists(st.integers(1, 8), min_size=1, max_size=3).map(tuple),
       dtype=st.sampled_from([torch.float32, torch.float64]),
       seed=st.integers(0, 2**31 - 1))
def test_allreduce_grad_identity(shape, dtype, seed):
    x = make(shape, dtype, seed).requires_grad_()
    g = make(shape, dtype, seed + 1)
    m.allreduce(x, m.SUM).backward(g)
    assert torch.equal(x.grad, g)


INT_DTYPES = [torch.int8, torch.int32, torch.int64, torch.uint8,
              torch.int16, torch.bool]


@settings(max_examples=40, deadline=None)
@given(shape=shapes, dtype=st.sampled_from(INT_DTYPES),
       op=st.sampled_from(["band", "bor", "bxor"]),
       seed=st.integers(0, 2**31 - 1))
def test_bitwise_size1_identity(shape, dtype, op, seed):
    """Bitwise reductions at size 1 are the identity for every integer
    dtype, and never mutate the input."""
    g = torch.Generator().manual_seed(seed)
    if dtype == torch.bool:
        x = torch.randint(0, 2, shape, generator=g).to(torch.bool)
    else:
        lo = 0 if dtype == torch.uint8 else -9
        x = torch.randint(lo, 9, shape, generator=g).to(dtype)
    ref = x.clone()
    o = m.Op(op)
    assert torch.equal(m.allreduce(x, o), x)
    assert torch.equal(m.scan(x, o), x)
    assert torch.equal(x, ref)


@settings(max_examples=20, deadline=None)
@given(dtype=st.sampled_from([torch.float32, torch.float16]),
       op=st.sampled_from(["band", "bor", "bxor"]))
def test_bitwise_rejects_float(dtype, op):
    import pytest as _pytest

    with _pytest.raises(ValueError):
        m.allreduce(torch.zeros(3, dtype=dtype), m.Op(op))
