"""Self-loopback exercise of every multi-rank native branch.

True multi-rank RCCL on ONE MI355X box is impossible on this pool — both
escape hatches were probed on hardware (profiles/multirank_evidence_r02.md):

* two ranks on one device: RCCL refuses at init
  (``NCCL WARN Duplicate GPU detected : rank 0 and rank 1 both on CUDA
  device 5d000``, rccl init.cc:1108 — gpurun_out/mrprobe2.log);
* CPX compute partitioning (8 logical devices): ``rocm-smi
  --setcomputepartition cpx`` is accepted but the partition stays SPX in
  this virtualized pool.

So these tests drive the identical native code paths with peer = self
inside one RCCL group — which RCCL executes through its normal transport
matching, not a shortcut of ours: the grouped send/recv composition
(bridge.cpp alltoall/gather/scatter pattern), the p2p chunk wrappers
across the 2^30-element boundary, the full ``sw_exchange``
pack → group → unpack pipeline (columns + interior rows + diagonal
corners, bridge.cpp:660-788), and the scan chain's recv+combine+send
sequence.  Cross-process matching of the same schedules is pinned by the
gloo multi-process suite (identical op-layer code, world 2/4) and by the
driver's round-end multi-GPU SCALE run.
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


def _ext():
    from mpi4jax_amd._backend import rccl

    return rccl.ext()


def _handle():
    from mpi4jax_amd.parallel.comm import get_default_comm

    return get_default_comm().rccl_handle()


def test_native_sendrecv_self_through_rccl():
    """bridge.cpp:416-430 with source=dest=self — real RCCL matching
    (the Python op layer short-circuits this; the native path must not
    need that shortcut)."""
    ext = _ext()
    x = torch.arange(4096, dtype=torch.float32, device="cuda")
    y = torch.empty_like(x)
    ext.sendrecv(x, y, 0, 0, _handle())
    torch.cuda.synchronize()
    assert torch.equal(x, y)


def test_native_grouped_send_recv_self():
    """Explicit group{send; recv} to self — the composition pattern every
    multi-rank grouped op uses (bridge.cpp:344-396)."""
    ext = _ext()
    h = _handle()
    x = torch.randn(1000, device="cuda")
    y = torch.empty_like(x)
    ext.group_start()
    ext.send(x, 0, h)
    ext.recv(y, 0, h)
    ext.group_end()
    torch.cuda.synchronize()
    assert torch.equal(x, y)


def test_p2p_chunk_boundary_self():
    """Crosses the kP2PChunk = 2^30-element boundary (bridge.cpp:304-326)
    so a transfer splits into two chunked send/recv pairs whose order must
    match on both sides."""
    ext = _ext()
    h = _handle()
    n = (1 << 30) + 4097
    x = torch.empty(n, dtype=torch.uint8, device="cuda")
    x[: 1 << 20].copy_(torch.arange(1 << 20, dtype=torch.int64) % 251)
    x[-4097:].copy_((torch.arange(4097, dtype=torch.int64) * 7) % 251)
    y = torch.empty_like(x)
    ext.sendrecv(x, y, 0, 0, h)
    torch.cuda.synchronize()
    assert torch.equal(x[: 1 << 20], y[: 1 << 20])
    assert torch.equal(x[-4097:], y[-4097:])
    del x, y
    torch.cuda.empty_cache()


@pytest.mark.parametrize("dtype", [torch.float32, torch.float64])
def test_sw_exchange_self_loopback_full_torus(dtype):
    """Drives EVERY remote branch of sw_exchange (bridge.cpp:660-788) in
    one call: two packed-column ops, both interior-row ops, all four
    diagonal corner ops — peers all self, i.e. a 1x1 fully-periodic
    torus expressed as remote transfers.  The result must equal the
    torus halo refresh computed in plain torch ("corners win" over
    column corner cells)."""
    ext = _ext()
    h = _handle()
    ny, nx = 37, 53
    nf = 3
    fields = [torch.randn(ny, nx, dtype=dtype, device="cuda")
              for _ in range(nf)]
    orig = [f.clone() for f in fields]

    col_bufs = [torch.empty(nf * ny, dtype=dtype, device="cuda")
                for _ in range(4)]
    cor_sbuf = torch.empty(4 * nf, dtype=dtype, device="cuda")
    cor_rbuf = torch.empty_like(cor_sbuf)

    # (k, send_to, recv_from, send_col, recv_col) — same encoding as
    # parallel/grid.halo_exchange_schedule
    col_ops = [0, 0, 0, 1, nx - 1,
               1, 0, 0, nx - 2, 0]
    # (send_to, recv_from, recv_row, send_row)
    row_ops = [0, 0, 0, ny - 2,
               0, 0, ny - 1, 1]
    # (d, send_to, recv_from) x 4
    cor_ops = [0, 0, 0, 1, 0, 0, 2, 0, 0, 3, 0, 0]
    cor_mask = 0b1111

    ext.sw_exchange(fields, [], col_ops, row_ops, cor_ops, cor_mask,
                    col_bufs, cor_sbuf, cor_rbuf, h)
    torch.cuda.synchronize()

    for f, x in zip(fields, orig):
        e = x.clone()
        e[:, nx - 1] = x[:, 1]
        e[:, 0] = x[:, nx - 2]
        e[0, 1:nx - 1] = x[ny - 2, 1:nx - 1]
        e[ny - 1, 1:nx - 1] = x[1, 1:nx - 1]
        e[ny - 1, nx - 1] = x[1, 1]          # corners win (d=0)
        e[ny - 1, 0] = x[1, nx - 2]          # d=1
        e[0, nx - 1] = x[ny - 2, 1]          # d=2
        e[0, 0] = x[ny - 2, nx - 2]          # d=3
        assert torch.equal(f, e)


def test_sw_exchange_partial_schedule_rows_only():
    """Row-only remote schedule (y-decomposition shape at dims=(2,1)):
    interior rows travel in-place, no column packing."""
    ext = _ext()
    h = _handle()
    ny, nx = 19, 23
    f = torch.randn(ny, nx, dtype=torch.float64, device="cuda")
    x = f.clone()
    empty = torch.empty(0, dtype=torch.float64, device="cuda")
    row_ops = [0, 0, 0, ny - 2,
               0, 0, ny - 1, 1]
    ext.sw_exchange([f], [], [], row_ops, [], 0,
                    [empty, empty, empty, empty], empty, empty, h)
    torch.cuda.synchronize()
    e = x.clone()
    e[0, 1:nx - 1] = x[ny - 2, 1:nx - 1]
    e[ny - 1, 1:nx - 1] = x[1, 1:nx - 1]
    assert torch.equal(f, e)


def test_envelope_stash_drain_device_order():
    """Device side of the envelope plane (_backend/envelope.py): when a
    recv wants the SECOND message in a peer's pipe, claim() must drain the
    first into a device stash with real RCCL recvs in send order.  The
    protocol itself is pinned at world 2 over gloo (tests/test_envelope.py);
    here the data plane is real RCCL with peer = self, envelopes
    pre-queued (the gloo leg carries no data, so skipping it changes
    nothing on the device side)."""
    from mpi4jax_amd._backend.envelope import EnvelopeBox
    from mpi4jax_amd.parallel.comm import get_default_comm

    ext = _ext()
    h = _handle()
    box = EnvelopeBox(get_default_comm())
    a = torch.full((16,), 7, dtype=torch.uint8, device="cuda")
    b = torch.arange(32, dtype=torch.uint8, device="cuda")

    drained = []

    def recv_bytes(src, nbytes):
        buf = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
        ext.recv(buf, src, h)
        drained.append((src, nbytes))
        return buf

    # all p2p with self must share one group (RCCL self-matching)
    ext.group_start()
    ext.send(a, 0, h)            # tag-7 message, first in the pipe
    ext.send(b, 0, h)            # tag-9 message, wanted first by the recv
    box.queue.append([0, 7, 16])
    box.queue.append([0, 9, 32])
    s, t, data = box.claim(0, 9, recv_bytes)
    assert (s, t, data) == (0, 9, None)
    assert drained == [(0, 16)], "tag-7 must drain (in send order) first"
    out = torch.empty(32, dtype=torch.uint8, device="cuda")
    ext.recv(out, 0, h)
    ext.group_end()
    torch.cuda.synchronize()

    assert torch.equal(out, b)
    s, t, data = box.claim(0, 7, recv_bytes)  # satisfied from the stash
    assert (s, t) == (0, 7) and data is not None
    torch.cuda.synchronize()
    assert torch.equal(data, a)
    assert box.queue == [] and box.stash == []


def test_scan_chain_sequence_self():
    """The scan ring's per-rank native sequence (bridge.cpp:446-468):
    recv the running prefix, combine on the CDNA4 kernel, send onward —
    driven with self as both neighbors inside a group."""
    ext = _ext()
    h = _handle()
    from mpi4jax_amd.ops.reduce_ops import Op, RCCL_OP_ENUM

    prefix = torch.randn(10000, device="cuda")  # "rank r-1's prefix"
    own = torch.randn(10000, device="cuda")
    got = torch.empty_like(prefix)
    ext.group_start()
    ext.send(prefix, 0, h)
    ext.recv(got, 0, h)
    ext.group_end()
    out = torch.empty_like(own)
    ext.combine(out, got, own, RCCL_OP_ENUM[Op.SUM])
    torch.cuda.synchronize()
    assert torch.allclose(out, prefix + own)
