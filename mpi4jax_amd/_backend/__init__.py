"""Backend dispatch: CPU (gloo bootstrap plane) vs GPU (native RCCL/HIP).

The reference dispatches per XLA platform with decorator-wrapped lowerings
(``/root/reference/mpi4jax/_src/decorators.py``).  Here dispatch is simply by
tensor device: host tensors ride the gloo bootstrap plane, device tensors go
through the hand-written HIP/RCCL extension.  There is no copy-to-host mode
and no multi-vendor dispatch — device traffic is always RCCL over xGMI.
"""

import torch


def is_gpu(x: torch.Tensor) -> bool:
    return x.is_cuda


def backend_for(x: torch.Tensor):
    if x.is_cuda:
        from . import rccl

        return rccl
    from . import cpu

    return cpu


def has_rccl_support() -> bool:
    """True when the native HIP/RCCL extension is importable."""
    try:
        from . import rccl

        rccl.ext()
        return True
    except Exception:
        return False


# capability-probe parity with the reference API
# (`mpi4jax.has_cuda_support`, utils.py:159-174) — for us "GPU support"
# means the RCCL extension.
def has_cuda_support() -> bool:
    """Alias of :func:`has_rccl_support` (drop-in parity with the
    reference's probe name, utils.py:159-166)."""
    return has_rccl_support()


def has_sycl_support() -> bool:
    """Always False — single-vendor by design (drop-in parity with the
    reference's capability probe, utils.py:159-174)."""
    return False


def version_info() -> dict:
    """RCCL / HIP runtime versions (diagnostic parity with the
    reference's MPI_ABI_INFO export, mpi_xla_bridge_cpu.cpp:524-533)."""
    try:
        from . import rccl

        return dict(rccl.ext().version_info())
    except Exception:
        return {}
