"""Bootstrap-plane message envelopes: MPI matching for order-matched
data planes.

Neither RCCL nor a single gloo wire tag has a message envelope: remote
send/recv match purely by enqueue order per (sender, receiver) pair.
That forbids MPI-style tag matching and ``recv(ANY_SOURCE)`` — the p2p
machinery the reference gets from MPI itself
(``mpi_ops_common.h:354-367``).  This plane restores it for both
backends: every remote send posts a tiny envelope ``(tag, nbytes)`` over
the communicator's gloo bootstrap group, and ``recv`` consumes envelopes
to decide WHICH data-plane message to take next.  The CPU backend rides
it unconditionally (``_backend/cpu.py``); the RCCL backend behind
``MPI4JAX_AMD_GPU_ENVELOPE=1`` (default off: the envelope costs ~10 µs
of host latency per message, and order-matched default-tag p2p — every
reference-style workload — needs none of it).  On the GPU path recv
decides:

* envelope matches in order → direct ``ncclRecv`` into the output;
* a differently-tagged message is ahead in the pipe → it is
  ``ncclRecv``-ed into a device stash buffer first (preserving RCCL's
  per-pair order matching) and handed to the recv that later asks for it;
* ``ANY_SOURCE`` → block on the envelope plane (``dist.recv`` from any
  rank), learn the source, then proceed as above.

The data plane stays 100 % RCCL-on-stream (stash drains are stream
-ordered device recvs; no host staging); only the 24-byte envelopes ride
gloo.  ``sendrecv`` keeps the grouped default-tag path (mixing a stash
drain into a send+recv group could reorder against the peer's group).
"""

import os
import weakref

import torch
import torch.distributed as dist

from ..utils.status import ANY_SOURCE, ANY_TAG

# reserved gloo tag for envelope traffic on the shared bootstrap group;
# user tags ≥ 2^24 would collide (documented in docs/sharp-bits.md)
ENV_TAG = (1 << 24) + 17

_BOXES = weakref.WeakKeyDictionary()


def enabled():
    return os.environ.get("MPI4JAX_AMD_GPU_ENVELOPE", "") == "1"


def box_for(comm):
    b = _BOXES.get(comm)
    if b is None:
        b = _BOXES[comm] = EnvelopeBox(comm)
    return b


class EnvelopeBox:
    """Per-communicator envelope state.

    ``queue`` holds received-but-unclaimed envelopes in arrival order
    (FIFO per source — gloo preserves per-pair ordering); ``stash`` holds
    messages whose data was drained ahead of their matching recv.
    """

    def __init__(self, comm):
        self.comm = comm
        self.queue = []   # [src_comm_rank, tag, nbytes]
        self.stash = []   # (src_comm_rank, tag, uint8 tensor)
        self._isends = []

    # ---------------------------------------------------------- sender
    def post(self, dest, tag, nbytes):
        env = torch.tensor([int(tag), int(nbytes)], dtype=torch.int64)
        w = dist.isend(env, dst=self.comm.global_rank(dest),
                       group=self.comm.gloo_group, tag=ENV_TAG)
        self._isends.append((w, env))  # keep the buffer alive
        self._isends = [(w, e) for (w, e) in self._isends
                        if not w.is_completed()]

    # --------------------------------------------------------- receiver
    def _recv_one(self):
        env = torch.empty(2, dtype=torch.int64)
        sender = dist.recv(env, src=None, group=self.comm.gloo_group,
                           tag=ENV_TAG)
        src = self.comm._ranks.index(sender)
        self.queue.append([src, int(env[0]), int(env[1])])

    def claim(self, source, tag, recv_bytes):
        """Match (source, tag) against stash and envelopes.

        ``recv_bytes(src, nbytes) -> uint8 tensor`` enqueues the RCCL
        recv for a message being drained ahead of order.  Returns
        ``(src, tag, data)`` where data is a stashed uint8 tensor or
        None (caller performs the direct recv itself, keeping its output
        allocation zero-copy).
        """
        while True:
            for i, (s, t, buf) in enumerate(self.stash):
                if source in (ANY_SOURCE, s) and tag in (ANY_TAG, t):
                    self.stash.pop(i)
                    return s, t, buf
            for i, (s, t, nb) in enumerate(self.queue):
                if source in (ANY_SOURCE, s) and tag in (ANY_TAG, t):
                    # drain every EARLIER message from the same source so
                    # the RCCL recv order per pair equals the send order
                    drain = [j for j in range(i) if self.queue[j][0] == s]
                    for j in drain:
                        es, et, enb = self.queue[j]
                        self.stash.append((es, et, recv_bytes(es, enb)))
                    for j in reversed(drain + [i]):
                        self.queue.pop(j)
                    return s, t, None
            self._recv_one()
