"""Watchdog: a wedged p2p must kill the job loudly.

RCCL enqueues asynchronously — a mismatched send/recv never errors, it
silently wedges the stream.  With MPI4JAX_AMD_WATCHDOG_SEC set, every
collective enqueue records a stream event; a monitor thread calls
ncclCommAbort on all communicators and exits (code 87, rank-tagged
stderr) when an event is still pending past the deadline — the
reference's abort discipline (mpi_ops_common.h:60-78) adapted to
stream-ordered enqueue.
"""

import os
import subprocess
import sys
import textwrap

import pytest
import torch

import mpi4jax_amd as m

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_watchdog_no_false_positive():
    """Healthy traffic with the watchdog armed must complete untouched."""
    from mpi4jax_amd._backend import rccl

    m.init()
    ext = rccl.ext()
    prev = ext.get_watchdog()
    ext.set_watchdog(30.0)
    try:
        x = torch.randn(1 << 20, device="cuda")
        for _ in range(5):
            y = m.allreduce(x, m.SUM)
        torch.cuda.synchronize()
        assert torch.equal(y, x)
    finally:
        ext.set_watchdog(prev)


WEDGE_SCRIPT = textwrap.dedent("""
    import os, sys, torch
    os.environ["MPI4JAX_AMD_WATCHDOG_SEC"] = "2"
    sys.path.insert(0, %r)
    import mpi4jax_amd as m
    m.init()
    from mpi4jax_amd._backend import rccl
    ext = rccl.ext()
    torch.zeros(4, device="cuda")  # init the device/stream
    # wedge the stream: a bounded 60s spin standing in for an unmatched
    # remote recv (world-1 RCCL self-mismatches error synchronously,
    # which their own test covers; real wedges need a peer)
    ext.debug_wedge_stream(60.0)
    print("WEDGE_ENQUEUED", flush=True)
    torch.cuda.synchronize()
    print("SYNC_COMPLETED_UNEXPECTEDLY", flush=True)
""")


def test_watchdog_kills_wedged_stream():
    r = subprocess.run(
        [sys.executable, "-c", WEDGE_SCRIPT % REPO], cwd=REPO,
        capture_output=True, text=True, timeout=120,
    )
    out, err = r.stdout, r.stderr
    assert "SYNC_COMPLETED_UNEXPECTEDLY" not in out, out
    assert r.returncode == 87, (r.returncode, out, err)
    assert "WATCHDOG" in err, err
    assert "still pending" in err, err


SELF_MISMATCH_SCRIPT = textwrap.dedent("""
    import os, sys, torch
    os.environ["MPI4JAX_AMD_WATCHDOG_SEC"] = "5"  # guard: wedge -> 87
    sys.path.insert(0, %r)
    import mpi4jax_amd as m
    m.init()
    from mpi4jax_amd._backend import rccl
    from mpi4jax_amd.parallel.comm import get_default_comm
    ext = rccl.ext()
    h = get_default_comm().rccl_handle()
    y = torch.empty(8, device="cuda")
    try:
        ext.recv(y, 0, h)
        torch.cuda.synchronize()
    except RuntimeError as e:
        print("FAILED_FAST:", str(e)[:80], flush=True)
        sys.exit(0)
    print("COMPLETED_SILENTLY", flush=True)
    sys.exit(3)
""")


def test_mismatched_self_p2p_fails_loudly():
    """An unmatched self recv must end loudly — either a synchronous RCCL
    error (observed on this runtime) or, failing that, the watchdog."""
    r = subprocess.run(
        [sys.executable, "-c", SELF_MISMATCH_SCRIPT % REPO], cwd=REPO,
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode in (0, 87), (r.returncode, r.stdout, r.stderr)
    assert "COMPLETED_SILENTLY" not in r.stdout
