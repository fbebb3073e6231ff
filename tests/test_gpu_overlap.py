"""Halo/compute overlap groundwork.

MPI4JAX_AMD_SW_OVERLAP=1 runs the h/u/v halo exchange on a second HIP
stream while the main stream computes the halo-independent friction pairs
(stage 28); the halo-dependent ring (stage 29) joins via events.  Stages
28+29 call the identical per-pair code as the serial stage 27, so the
overlapped step must be BITWISE equal to the serial one.  The
``_force_remote_exchange`` hook expresses the world-1 periodic wraps as
RCCL self-transfers so the comm stream carries real grouped p2p here —
the stream-discipline validation at N=1 the round-1 verdict asked for.
"""

import os

import pytest
import torch

import mpi4jax_amd as m
from mpi4jax_amd.models import ShallowWater

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    m.init()
    yield
    torch.cuda.synchronize()


def _traj(monkeypatch_env, steps=20, force_remote=False):
    for k, v in monkeypatch_env.items():
        if v is None:
            os.environ.pop(k, None)
        else:
            os.environ[k] = v
    try:
        torch.manual_seed(0)
        sw = ShallowWater(nx=256, ny=128, device="cuda",
                          dtype=torch.float32,
                          _force_remote_exchange=force_remote)
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(steps):
            s = sw.step(s)
        torch.cuda.synchronize()
        return sw, {k: getattr(s, k).clone() for k in ("h", "u", "v")}
    finally:
        for k in monkeypatch_env:
            os.environ.pop(k, None)


def test_overlap_bitwise_equal_serial():
    # serial baseline = the two-kernel 19+27 pipeline (28+29 reuse its
    # pair code); stage 30, the world-1 default, fuses differently
    _, base = _traj({"MPI4JAX_AMD_SW_OVERLAP": None,
                     "MPI4JAX_AMD_SW_NOFUSE": "1"})
    sw, ov = _traj({"MPI4JAX_AMD_SW_OVERLAP": "1"})
    assert sw._overlap_plan() is not None, "overlap path not active"
    for k in ("h", "u", "v"):
        assert torch.equal(base[k], ov[k]), k


def test_overlap_with_forced_remote_exchange():
    """Comm stream carries real RCCL self-transfers (pack, grouped p2p,
    unpack) while stage 28 runs on the main stream."""
    _, base = _traj({"MPI4JAX_AMD_SW_OVERLAP": None,
                     "MPI4JAX_AMD_SW_NOFUSE": "1"})
    sw, ov = _traj({"MPI4JAX_AMD_SW_OVERLAP": "1"}, force_remote=True)
    assert sw._overlap_plan() is not None
    assert sw._exchange_cache()["comm_id"] != -1, "RCCL path not used"
    for k in ("h", "u", "v"):
        assert torch.equal(base[k], ov[k]), k


def test_overlap_disables_graph_capture():
    """hipGraph capture of the two-stream overlapped step segfaults in
    the HIP runtime on ROCm 7.0 (observed: SIGSEGV inside capture, not a
    catchable error — gpurun_out/ov logs).  make_stepper must therefore
    fall back to the eager loop when the overlap path is active, and the
    stepper must still advance correctly."""
    os.environ["MPI4JAX_AMD_SW_OVERLAP"] = "1"
    try:
        torch.manual_seed(0)
        sw = ShallowWater(nx=256, ny=128, device="cuda",
                          dtype=torch.float32,
                          _force_remote_exchange=True)
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(3):
            s = sw.step(s)
        advance, s = sw.make_stepper(s, steps_per_call=4)
        for _ in range(5):
            s = advance()
        torch.cuda.synchronize()
        assert torch.isfinite(s.h).all()
    finally:
        os.environ.pop("MPI4JAX_AMD_SW_OVERLAP", None)
