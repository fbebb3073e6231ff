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


# ------------------------------------------- build/runtime compat guard

def test_build_compat_guard(monkeypatch):
    """Runtime-vs-build guard (reference MPI ABI check analog): torch
    major.minor mismatch and RCCL major mismatch raise; minor RCCL skew
    only warns; env bypass works."""
    import warnings

    import torch

    from mpi4jax_amd._backend.rccl import _check_build_compat

    cur = torch.__version__.split("+")[0]

    class Ext:
        def __init__(self, t=cur, bh=22707, rt=22707):
            self._t, self._bh, self._rt = t, bh, rt

        def build_info(self):
            return {"torch": self._t, "rccl_header": self._bh,
                    "glibcxx_use_cxx11_abi": 1}

        def version_info(self):
            return {"rccl": self._rt, "hip_runtime": 0}

    monkeypatch.delenv("MPI4JAX_AMD_SKIP_ABI_CHECK", raising=False)
    _check_build_compat(Ext())  # exact match: fine
    with pytest.raises(ImportError, match="C\\+\\+ ABI"):
        _check_build_compat(Ext(t="9.9.0"))
    with pytest.raises(ImportError, match="major-version"):
        _check_build_compat(Ext(bh=32707))
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        _check_build_compat(Ext(rt=22606))  # the torch-lib librccl skew
        assert any("skew" in str(x.message) for x in w)
    monkeypatch.setenv("MPI4JAX_AMD_SKIP_ABI_CHECK", "1")
    _check_build_compat(Ext(t="9.9.0"))  # bypassed
