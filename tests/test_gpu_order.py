"""Cross-stream collective ordering (the ordered-effect analog).

The reference pinned "program order == network order" with a JAX ordered
effect threaded through every primitive
(``/root/reference/mpi4jax/_src/utils.py:45-53``); the torch analog of the
hazard it closed is collectives enqueued on DIFFERENT HIP streams racing
in submission order.  ``_backend/rccl._order_fence`` bridges each
collective's stream to the previous collective's stream with an event, so
per-communicator program order is the network order even across streams —
no user synchronization required.  Verified here by wedging stream A with
a spin kernel so that, without the fence, stream B's collective would
read A's result buffer long before it is written.
"""

import pytest
import torch

import mpi4jax_amd as m

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    m.init()
    yield
    torch.cuda.synchronize()


def test_cross_stream_program_order_is_network_order():
    from mpi4jax_amd._backend import rccl

    s_a = torch.cuda.Stream()
    s_b = torch.cuda.Stream()
    with torch.cuda.stream(s_a):
        rccl.ext().debug_wedge_stream(0.5)  # stall A's collective
        x = torch.full((1 << 20,), 3.0, device="cuda")
        y = m.allreduce(x, m.Op.SUM)  # enqueued on A, behind the wedge
    with torch.cuda.stream(s_b):
        # without the fence this reads y's buffer ~0.5 s before A's
        # allreduce writes it
        z = m.allreduce(y, m.Op.SUM)
        z.record_stream(s_b)
    torch.cuda.synchronize()
    ws = m.get_world().size
    assert torch.equal(z, torch.full_like(z, 3.0 * ws * ws))


def test_same_stream_fence_is_noop_and_state_tracks():
    from mpi4jax_amd._backend.rccl import _ORDER_STREAMS
    from mpi4jax_amd.parallel.comm import get_default_comm

    x = torch.ones(64, device="cuda")
    m.allreduce(x, m.Op.SUM)
    comm = get_default_comm()
    assert _ORDER_STREAMS.get(comm) == torch.cuda.current_stream()
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        m.allreduce(x.to("cuda"), m.Op.SUM)
    assert _ORDER_STREAMS.get(comm) == s
    m.allreduce(x, m.Op.SUM)  # bridge back to the default stream
    torch.cuda.synchronize()
