"""Debug logger with the reference's line format.

The reference wraps every MPI call in an RAII ``DebugTimer`` that prints
``r<rank> | <8-char id> | MPI_<Op> ...`` before and
``r<rank> | <id> | done with code 0 (x.xxe-ys)`` after each call
(``/root/reference/mpi4jax/_src/xla_bridge/mpi_ops_common.h:154-206``),
toggled by ``MPI4JAX_DEBUG`` (``xla_bridge/__init__.py:114-129``).

Here the toggle is ``MPI4JAX_AMD_DEBUG`` and the logged duration is the
*enqueue* time on the GPU path (collectives are asynchronous on the HIP
stream by design; a blocking timer would destroy the zero-sync property).
The CPU (gloo) path logs true call duration.
"""

import os
import random
import string
import time
from contextlib import contextmanager

_LOGGING = False


def set_logging(enabled: bool):
    """Toggle debug logging on every backend (reference:
    xla_bridge/__init__.py:114-129)."""
    global _LOGGING
    _LOGGING = bool(enabled)
    # keep the native extension's flag in sync if it is loaded
    try:
        from .._backend import rccl

        if rccl.ext_is_loaded():
            rccl.ext().set_logging(_LOGGING)
    except Exception:
        pass


def get_logging() -> bool:
    """Current debug-logging state."""
    return _LOGGING


def _random_id(n=8):
    return "".join(random.choices(string.ascii_lowercase + string.digits, k=n))


@contextmanager
def debug_timer(op_name: str, rank: int, detail: str = ""):
    """Context manager printing the reference-format debug lines."""
    if not _LOGGING:
        yield
        return
    uid = _random_id()
    suffix = f" ({detail})" if detail else ""
    print(f"r{rank} | {uid} | {op_name}{suffix}", flush=True)
    t0 = time.perf_counter()
    yield
    dt = time.perf_counter() - t0
    print(f"r{rank} | {uid} | done with code 0 ({dt:.2e}s)", flush=True)


# init from env at import, like the reference (xla_bridge/__init__.py:128-129)
if os.environ.get("MPI4JAX_AMD_DEBUG", "").strip() not in ("", "0", "false", "False"):
    _LOGGING = True
