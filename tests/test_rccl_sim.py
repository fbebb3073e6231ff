"""World-2 exercise of the REAL `_backend/rccl.py` p2p code paths.

True multi-rank RCCL cannot run on this pool (one GPU per box; RCCL
refuses two ranks on one device — profiles/multirank_evidence_r02.md), so
the envelope-integrated send/recv code in `_backend/rccl.py:272-330` is
driven here with its native extension replaced by a shim whose matching
semantics are exactly RCCL's: FIFO per ordered (sender, receiver) pair,
no tags on the wire, sends that never block the host (RCCL enqueues on
the stream).  Everything else — the envelope plane over the real gloo
bootstrap group, the stash drain, Status synthesis, the ANY_SOURCE
resolution — is the production code, byte for byte.
"""

import os

import pytest
import torch
import torch.distributed as dist

from tests._mp import run_multiproc

DATA_TAG = (1 << 24) + 99  # distinct from envelope.ENV_TAG


class _FakeRcclExt:
    """RCCL-shaped data plane over gloo: order-matched, tag-blind,
    non-blocking sends."""

    def __init__(self, comm):
        self.comm = comm
        self.pending = []

    def send(self, buf, dest, handle):
        w = dist.isend(buf, dst=self.comm.global_rank(dest),
                       group=self.comm.gloo_group, tag=DATA_TAG)
        self.pending.append((w, buf))

    def recv(self, buf, src, handle):
        dist.recv(buf, src=self.comm.global_rank(src),
                  group=self.comm.gloo_group, tag=DATA_TAG)

    def flush(self):
        for w, _ in self.pending:
            w.wait()
        self.pending.clear()

    def check_async_errors(self):  # called by m.finalize()
        pass


def _sim_worker(rank, ws):
    os.environ["MPI4JAX_AMD_GPU_ENVELOPE"] = "1"
    import mpi4jax_amd as m
    from mpi4jax_amd._backend import rccl
    from mpi4jax_amd.utils.status import ANY_SOURCE, ANY_TAG, Status

    comm = m.get_world()
    fake = _FakeRcclExt(comm)
    rccl._EXT = fake                      # ext() returns the shim
    rccl._handle = lambda c: 0            # no RCCL init / CUDA fence

    t = torch.float64
    if rank == 0:
        rccl.send(torch.full((5,), 1.5, dtype=t), 1, 7, comm)
        rccl.send(torch.full((3, 2), 2.5, dtype=t), 1, 9, comm)
        rccl.send(torch.arange(4, dtype=t), 1, 0, comm)
        fake.flush()
    else:
        # 1. want tag 9 FIRST → the real recv() must drain the tag-7
        #    message into the stash with an order-preserving data recv
        st = Status()
        out = rccl.recv(torch.empty(3, 2, dtype=t), 0, 9, comm, st)
        assert torch.equal(out, torch.full((3, 2), 2.5, dtype=t))
        assert (st.source, st.tag, st.count) == (0, 9, 48)
        # 2. ANY_SOURCE/ANY_TAG picks up the stashed tag-7 message with
        #    no further data-plane traffic and synthesizes its envelope
        st = Status()
        out = rccl.recv(torch.empty(5, dtype=t), ANY_SOURCE, ANY_TAG,
                        comm, st)
        assert torch.equal(out, torch.full((5,), 1.5, dtype=t))
        assert (st.source, st.tag) == (0, 7)
        # 3. plain default-tag recv still flows through the envelope
        st = Status()
        out = rccl.recv(torch.empty(4, dtype=t), 0, 0, comm, st)
        assert torch.equal(out, torch.arange(4, dtype=t))
        assert (st.source, st.tag) == (0, 0)
        from mpi4jax_amd._backend.envelope import box_for

        box = box_for(comm)
        assert box.queue == [] and box.stash == []
    dist.barrier(group=comm.gloo_group)


def test_rccl_send_recv_envelope_world2():
    run_multiproc(_sim_worker, 2)


def _strict_worker(rank, ws):
    """Without the envelope plane the production code must fail loudly —
    never mismatch silently."""
    os.environ.pop("MPI4JAX_AMD_GPU_ENVELOPE", None)
    os.environ.pop("MPI4JAX_AMD_ALLOW_GPU_TAGS", None)
    import mpi4jax_amd as m
    from mpi4jax_amd._backend import rccl
    from mpi4jax_amd.utils.status import ANY_SOURCE

    comm = m.get_world()
    with pytest.raises(ValueError, match="tag"):
        rccl.send(torch.ones(3), 1 - rank, 5, comm)
    with pytest.raises(ValueError, match="ANY_SOURCE"):
        rccl.recv(torch.empty(3), ANY_SOURCE, 0, comm, None)


def test_gpu_tags_fail_loudly_without_envelope():
    run_multiproc(_strict_worker, 2)


def _random_worker(rank, ws, seed):
    """Property test: random tags/sizes/recv strategies at world 2 must
    follow MPI matching — a recv by (source, tag) delivers the
    EARLIEST-sent unconsumed matching message; ANY_TAG/ANY_SOURCE
    deliver the earliest unconsumed message outright (the stash is
    always older than the queue, so claim()'s stash-first order is the
    send order)."""
    import random

    import mpi4jax_amd as m
    from mpi4jax_amd._backend import rccl
    from mpi4jax_amd.utils.status import ANY_SOURCE, ANY_TAG, Status

    os.environ["MPI4JAX_AMD_GPU_ENVELOPE"] = "1"
    comm = m.get_world()
    fake = _FakeRcclExt(comm)
    rccl._EXT = fake
    rccl._handle = lambda c: 0

    rng = random.Random(seed)
    n_msgs = 14
    msgs = [(k, rng.randrange(0, 4), rng.randrange(1, 9))
            for k in range(n_msgs)]  # (serial, tag, numel)
    if rank == 0:
        for serial, tag, numel in msgs:
            rccl.send(torch.full((numel,), float(serial)), 1, tag, comm)
        fake.flush()
    else:
        remaining = list(msgs)
        while remaining:
            mode = rng.randrange(3)
            if mode == 0:          # specific tag (of a remaining msg)
                tag = rng.choice(remaining)[1]
                src = 0
            elif mode == 1:        # ANY_TAG
                tag, src = ANY_TAG, 0
            else:                  # full wildcard
                tag, src = ANY_TAG, ANY_SOURCE
            exp = next(mm for mm in remaining
                       if tag in (ANY_TAG, mm[1]))
            remaining.remove(exp)
            st = Status()
            out = rccl.recv(torch.empty(exp[2]), src, tag, comm, st)
            assert torch.equal(out, torch.full((exp[2],),
                                               float(exp[0]))), (
                exp, out[0].item())
            assert (st.source, st.tag) == (0, exp[1])
        from mpi4jax_amd._backend.envelope import box_for

        box = box_for(comm)
        assert box.queue == [] and box.stash == []
    dist.barrier(group=comm.gloo_group)


@pytest.mark.parametrize("seed", [7, 23, 1009])
def test_rccl_envelope_random_matching_world2(seed):
    run_multiproc(_random_worker, 2, args=(seed,))
