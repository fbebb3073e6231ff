import pytest
import torch

import mpi4jax_amd as m
from mpi4jax_amd.utils.validation import enforce_types


def test_enforce_types_basic():
    @enforce_types(root=int, name=str)
    def f(x, root, name="a"):
        return root

    assert f(1, 2) == 2
    assert f(1, root=3, name="b") == 3
    with pytest.raises(TypeError, match="root"):
        f(1, "zero")


def test_enforce_types_numpy_int():
    import numpy as np

    @enforce_types(root=int)
    def f(root):
        return root

    assert f(np.int32(2)) == 2


def test_tensor_as_static_arg():
    # analog of the reference's tracer-leak message (validation.py:77-88)
    with pytest.raises(TypeError, match="static"):
        m.bcast(torch.zeros(2), torch.tensor(0))


def test_bad_comm_type():
    with pytest.raises(TypeError, match="comm"):
        m.allreduce(torch.zeros(2), m.SUM, comm="world")


def test_unsupported_dtype():
    # no uint32 in the supported table
    with pytest.raises(TypeError, match="dtype"):
        m.allreduce(torch.zeros(2, dtype=torch.uint32), m.SUM)


def test_scan_avg_rejected():
    with pytest.raises(ValueError):
        m.scan(torch.zeros(2), m.AVG)


def test_sendrecv_device_mismatch():
    if not torch.cuda.is_available():
        import pytest as _pytest

        _pytest.skip("needs a GPU to construct the mismatch")
    with pytest.raises(ValueError, match="same device"):
        m.sendrecv(torch.zeros(3), torch.zeros(3, device="cuda"),
                   source=0, dest=0)


def test_string_op_names():
    """Eager ops accept string op names, same spelling as jit_ops."""
    x = torch.ones(3)
    assert torch.equal(m.allreduce(x, "sum"), x)
    assert torch.equal(m.scan(x, "MAX"), x)
    with pytest.raises(TypeError):
        m.allreduce(x, "not-an-op")
