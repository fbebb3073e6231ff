"""mpi4jax_amd — MI355X-native zero-copy collectives on PyTorch-ROCm tensors.

A brand-new framework with the capabilities of mpi4jax (reference public API:
``/root/reference/mpi4jax/__init__.py:8-41``): the twelve communication
primitives callable on device tensors with zero-copy semantics, ordering
guarantees, and autodiff rules for ``allreduce(SUM)`` and ``sendrecv``.

MI355X-native design (NOT a port):

* compute substrate is PyTorch-ROCm tensors, not JAX arrays — there is no
  XLA; ordering comes from HIP stream order (all collectives are enqueued on
  the *current* compute stream with zero synchronization and zero staging
  copies), with an optional ``token`` kwarg kept for API parity that rejects
  explicit tokens exactly like the reference (``_src/utils.py:30-42``);
* the GPU data path is a hand-written C++/HIP extension that launches RCCL
  collectives directly on the current ``hipStream_t`` over xGMI — the
  reference's stream-synchronize + host-MPI path and its copy-to-host mode
  (``mpi_xla_bridge_cuda.cpp:185-206``) do not exist here;
* send/recv/sendrecv/alltoall/gather/scatter lower to grouped
  ``ncclSend``/``ncclRecv``; scan is a ring with an on-device CDNA4 combine
  kernel;
* the CPU path (for GPU-less correctness testing and host arrays) uses the
  gloo process group of ``torch.distributed``.
"""

from ._version import __version__  # noqa: F401

from .utils.tokens import NOTSET, Token  # noqa: F401
from .ops.reduce_ops import (  # noqa: F401
    Op,
    SUM,
    PROD,
    MIN,
    MAX,
    AVG,
    BAND,
    BOR,
    BXOR,
)
from .utils.status import Status, ANY_SOURCE, ANY_TAG  # noqa: F401
from .parallel.comm import (  # noqa: F401
    Communicator,
    init,
    finalize,
    get_default_comm,
    get_world,
    COMM_WORLD,
)
from ._backend import (  # noqa: F401
    has_rccl_support,
    has_cuda_support,
    has_sycl_support,
    version_info,
)
from .utils.logging import set_logging, get_logging  # noqa: F401

from .ops import jit_ops  # noqa: F401

# make `import mpi4jax_amd.jit_ops` work too (it lives under ops/)
import sys as _sys

_sys.modules[__name__ + ".jit_ops"] = jit_ops
del _sys
from .ops import (  # noqa: F401
    allgather,
    allreduce,
    alltoall,
    barrier,
    bcast,
    gather,
    recv,
    reduce,
    reduce_scatter,
    scan,
    scatter,
    send,
    sendrecv,
)

__all__ = [
    "allgather",
    "allreduce",
    "alltoall",
    "barrier",
    "bcast",
    "gather",
    "recv",
    "reduce",
    "reduce_scatter",
    "scan",
    "scatter",
    "send",
    "sendrecv",
    "has_rccl_support",
    "has_cuda_support",
    "has_sycl_support",
    "version_info",
    "Op",
    "SUM",
    "PROD",
    "MIN",
    "MAX",
    "AVG",
    "BAND",
    "BOR",
    "BXOR",
    "Status",
    "ANY_SOURCE",
    "ANY_TAG",
    "Communicator",
    "init",
    "finalize",
    "get_default_comm",
    "COMM_WORLD",
    "jit_ops",
    "set_logging",
    "get_logging",
    "NOTSET",
    "Token",
]

# Exit-time flush: the reference registers an atexit hook that runs
# jax.effects_barrier() so pending async communication cannot deadlock the
# interpreter at exit (_src/__init__.py:14-24).  Our analog synchronizes the
# device so all enqueued RCCL work drains before Python tears down.
import atexit as _atexit  # noqa: E402
from .parallel.comm import flush  # noqa: E402,F401

_atexit.register(flush)
