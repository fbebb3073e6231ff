"""Envelope plane for GPU p2p tag matching / ANY_SOURCE (CPU-testable).

The RCCL data plane matches messages by enqueue order only; the envelope
plane (``_backend/envelope.py``) adds MPI-style matching on top.  The
claim algorithm (ordering, stash-ahead draining, ANY_SOURCE/ANY_TAG
wildcards) is transport-independent and pinned here without a GPU; the
world-2 test runs the full protocol over a real gloo group with a
byte-wire data plane standing in for RCCL (identical ordering semantics:
FIFO per ordered pair)."""

import torch

from mpi4jax_amd._backend.envelope import ENV_TAG, EnvelopeBox
from mpi4jax_amd.utils.status import ANY_SOURCE, ANY_TAG
from tests._mp import run_multiproc


class _FakeComm:
    def __init__(self, rank, size):
        self.rank, self.size = rank, size
        self.gloo_group = None
        self._ranks = list(range(size))

    def global_rank(self, r):
        return r


def _box(rank=0, size=4):
    return EnvelopeBox(_FakeComm(rank, size))


def _recv_recorder(log):
    def recv_bytes(src, nbytes):
        buf = torch.full((nbytes,), src * 16 + len(log),
                         dtype=torch.uint8)
        log.append((src, nbytes))
        return buf

    return recv_bytes


def test_claim_in_order_direct():
    b = _box()
    b.queue.append([2, 7, 64])
    log = []
    s, t, data = b.claim(2, 7, _recv_recorder(log))
    assert (s, t, data) == (2, 7, None)
    assert log == [] and b.queue == [] and b.stash == []


def test_claim_out_of_order_drains_earlier_same_source():
    b = _box()
    # source 1 sent tag 7 (first) then tag 9; recv wants tag 9 first
    b.queue.append([1, 7, 16])
    b.queue.append([1, 9, 32])
    log = []
    s, t, data = b.claim(1, 9, _recv_recorder(log))
    assert (s, t, data) == (1, 9, None)
    assert log == [(1, 16)], "tag-7 message must drain first (send order)"
    assert len(b.stash) == 1 and b.stash[0][:2] == (1, 7)
    # the stashed message satisfies the later recv without data-plane work
    log2 = []
    s, t, data = b.claim(1, 7, _recv_recorder(log2))
    assert s == 1 and t == 7 and data is not None and log2 == []
    assert b.stash == []


def test_claim_does_not_drain_other_sources():
    b = _box()
    b.queue.append([0, 5, 8])   # from source 0 — unrelated
    b.queue.append([3, 5, 8])
    log = []
    s, t, data = b.claim(3, 5, _recv_recorder(log))
    assert (s, t) == (3, 5) and data is None and log == []
    assert b.queue == [[0, 5, 8]]


def test_claim_wildcards():
    b = _box()
    b.queue.append([2, 11, 8])
    s, t, _ = b.claim(ANY_SOURCE, ANY_TAG, _recv_recorder([]))
    assert (s, t) == (2, 11)
    b.queue.append([1, 3, 8])
    b.queue.append([2, 4, 8])
    s, t, _ = b.claim(2, ANY_TAG, _recv_recorder([]))
    assert (s, t) == (2, 4)
    assert b.queue == [[1, 3, 8]]


def _proto_worker(rank, ws):
    """Full protocol over a real gloo group; byte-wire data plane."""
    import torch.distributed as dist

    import mpi4jax_amd as m

    comm = m.get_world()
    box = EnvelopeBox(comm)
    pending = []  # keep isend works + buffers alive (models the async
    #               RCCL data plane: the GPU host never blocks in send)

    def env_send(x, dest, tag):
        box.post(dest, tag, x.numel())
        w = dist.isend(x, dst=comm.global_rank(dest),
                       group=comm.gloo_group, tag=ENV_TAG + 1)
        pending.append((w, x))

    def recv_bytes(src, nbytes):
        buf = torch.empty(nbytes, dtype=torch.uint8)
        dist.recv(buf, src=comm.global_rank(src), group=comm.gloo_group,
                  tag=ENV_TAG + 1)
        return buf

    def env_recv(n, source, tag):
        s, t, data = box.claim(source, tag, recv_bytes)
        if data is None:
            data = recv_bytes(s, n)
        return s, t, data

    if rank == 0:
        env_send(torch.full((4,), 77, dtype=torch.uint8), 1, tag=7)
        env_send(torch.full((8,), 99, dtype=torch.uint8), 1, tag=9)
        env_send(torch.full((2,), 55, dtype=torch.uint8), 1, tag=0)
        for w, _ in pending:
            w.wait()
    else:
        # consume out of send order: 9 first (forces a stash of tag 7),
        # then ANY_SOURCE (matches the stashed 7), then exact (0)
        s, t, d = env_recv(8, 0, 9)
        assert (s, t) == (0, 9) and d[0].item() == 99, (s, t, d)
        s, t, d = env_recv(4, ANY_SOURCE, ANY_TAG)
        assert (s, t) == (0, 7) and d[0].item() == 77, (s, t, d)
        s, t, d = env_recv(2, 0, 0)
        assert (s, t) == (0, 0) and d[0].item() == 55, (s, t, d)
        assert box.stash == [] and box.queue == []


def test_envelope_protocol_world2_gloo():
    run_multiproc(_proto_worker, 2)


# ------------------------------------------------- property-based model
try:
    from hypothesis import given, settings, strategies as st

    HAVE_HYPOTHESIS = True
except ImportError:  # pragma: no cover
    HAVE_HYPOTHESIS = False


if HAVE_HYPOTHESIS:
    @settings(max_examples=200, deadline=None)
    @given(st.data())
    def test_claim_matches_mpi_model(data):
        """EnvelopeBox.claim vs MPI's guarantees: for any interleaving
        of arrivals and (source, tag) queries, the claimed message must
        match the query, be that SOURCE's earliest unconsumed matching
        message (per-pair non-overtaking — MPI does not order across
        sources), and data drains per source must follow send order;
        nothing may be left once everything is consumed."""
        n_src = data.draw(st.integers(1, 3))
        sends = data.draw(st.lists(
            st.tuples(st.integers(0, n_src - 1), st.integers(0, 3),
                      st.integers(1, 5)),
            min_size=1, max_size=12))
        box = _box(rank=3, size=4)
        # sent[k] = (src, tag, nbytes); serial k encodes payload
        pending = list(enumerate(sends))  # model: unconsumed, send order
        arrived = 0
        drains = {s: [] for s in range(n_src)}

        def recv_bytes(src, nbytes):
            drains[src].append(nbytes)
            return torch.full((nbytes,), 0, dtype=torch.uint8)

        while pending:
            # deliver a random number of envelopes ahead of the query
            k = data.draw(st.integers(0, len(sends) - arrived))
            for _ in range(k):
                s, t, nb = sends[arrived]
                box.queue.append([s, t, nb])
                arrived += 1
            # query a message that is guaranteed claimable (its envelope
            # either arrived or claim would block in _recv_one)
            claimable = [e for e in pending if e[0] < arrived]
            if not claimable:
                s, t, nb = sends[arrived]
                box.queue.append([s, t, nb])
                arrived += 1
                claimable = [e for e in pending if e[0] < arrived]
            serial, (esrc, etag, enb) = data.draw(st.sampled_from(claimable))
            use_any_src = data.draw(st.booleans())
            use_any_tag = data.draw(st.booleans())
            q_src = ANY_SOURCE if use_any_src else esrc
            q_tag = ANY_TAG if use_any_tag else etag
            got_s, got_t, _ = box.claim(q_src, q_tag, recv_bytes)
            # 1. the claim matches the query
            assert q_src in (ANY_SOURCE, got_s)
            assert q_tag in (ANY_TAG, got_t)
            # 2. non-overtaking: it is the claimed SOURCE's earliest
            #    unconsumed matching message
            exp = next(e for e in pending if e[1][0] == got_s
                       and (q_tag in (ANY_TAG, e[1][1])))
            assert got_t == exp[1][1], (sends, exp, (got_s, got_t))
            pending.remove(exp)
        assert box.queue == []
        # every stash entry must eventually have been claimed
        assert box.stash == []
        # drained messages follow send order per source (a subsequence
        # of that source's nbytes sequence, in order)
        for s in range(n_src):
            sent_nb = [nb for (src, _t, nb) in sends if src == s]
            it = iter(sent_nb)
            assert all(any(nb == x for x in it) for nb in drains[s]), (
                sends, s, drains[s])
