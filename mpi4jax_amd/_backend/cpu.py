"""CPU transport over the gloo process group.

Functional equivalent of the reference's CPU FFI backend
(``/root/reference/mpi4jax/_src/xla_bridge/mpi_xla_bridge_cpu.cpp``), used
for host tensors and for GPU-less correctness testing ("a cluster of one" —
SURVEY.md §4).  All functions take comm-relative ranks and NEVER mutate
their inputs (the reference's immutability contract).
"""

import weakref

import torch
import torch.distributed as dist

from . import envelope
from ..ops.reduce_ops import Op, GLOO_OP_MAP, combine
from ..utils.status import ANY_SOURCE, ANY_TAG
from ..utils.logging import debug_timer

# Per-communicator FIFO queues for same-rank send->recv (self-messaging).
# MPI supports buffered self sends; gloo does not, so we emulate locally.
# Weak-keyed so a GC'd communicator can never alias a new one's queue.
_SELF_QUEUES = weakref.WeakKeyDictionary()

# Remote p2p rides the SAME envelope plane as the RCCL backend
# (``_backend/envelope.py``): data travels on one fixed wire tag (FIFO
# per ordered pair — exactly RCCL's matching), envelopes carry
# ``(tag, nbytes)``, and ``claim()`` provides full MPI matching —
# earliest-match by (source, tag), ANY_SOURCE/ANY_TAG wildcards, and
# Status synthesis from the envelope (gloo's own tag matching cannot
# express ANY_TAG, and it reorders delivery relative to MPI's
# earliest-send rule).  Data sends are non-blocking so a differently-
# tagged message ahead in the pipe can never head-block the sender.
WIRE_TAG = envelope.ENV_TAG + 1

_PENDING = weakref.WeakKeyDictionary()  # comm -> [(isend work, buffer)]


def _isend_data(comm, wire, dest):
    w = dist.isend(wire, dst=comm.global_rank(dest),
                   group=comm.gloo_group, tag=WIRE_TAG)
    pl = _PENDING.setdefault(comm, [])
    pl.append((w, wire))
    pl[:] = [(ww, b) for ww, b in pl if not ww.is_completed()]


def drain_pending(timeout_s=60.0):
    """Wait for all in-flight data sends (MPI_Finalize semantics: every
    send must be matched and delivered).  Bounded: an UNMATCHED send at
    finalize is a bug in the program — after ``timeout_s`` this warns
    loudly and abandons the sends instead of hanging interpreter exit
    (MPI implementations typically abort here)."""
    import time
    import warnings

    deadline = time.monotonic() + timeout_s
    leftover = 0
    for pl in _PENDING.values():
        for w, _ in pl:
            while not w.is_completed():
                if time.monotonic() > deadline:
                    leftover += 1
                    break
                time.sleep(0.005)
        pl.clear()
    if leftover:
        warnings.warn(
            f"mpi4jax_amd.finalize: {leftover} point-to-point send(s) "
            "were never matched by a receive — abandoned after "
            f"{timeout_s:.0f}s (check your send/recv pairing)",
            stacklevel=2,
        )


def _env_recv_into(out, source, tag, comm, status):
    def recv_bytes(src, nbytes):
        buf = torch.empty(nbytes, dtype=torch.uint8)
        dist.recv(buf, src=comm.global_rank(src), group=comm.gloo_group,
                  tag=WIRE_TAG)
        return buf

    s, t, data = envelope.box_for(comm).claim(source, tag, recv_bytes)
    if data is not None:  # drained ahead of order into the stash
        out.copy_(data.view(out.dtype).reshape(out.shape))
    else:
        dist.recv(_wire(out), src=comm.global_rank(s),
                  group=comm.gloo_group, tag=WIRE_TAG)
    _fill_status(status, s, t, out)
    return out


def _self_queue(comm):
    return _SELF_QUEUES.setdefault(comm, [])


def _wire(t):
    """1-D uint8 view sharing storage — gloo supports every op on bytes,
    so exotic dtypes (bf16/f16/int16/complex/bool) travel unchanged."""
    return t.view(torch.uint8) if t.dtype == torch.uint8 else \
        t.contiguous().flatten().view(torch.uint8)


_BITWISE_OPS = (Op.BAND, Op.BOR, Op.BXOR)


def _check_bitwise_dtype(op, x, op_name):
    if op in _BITWISE_OPS and (x.is_floating_point() or x.is_complex()):
        raise ValueError(
            f"{op_name}: {op} requires an integer or bool dtype, "
            f"got {x.dtype}"
        )


def _gloo_reduce_op(op: Op, dtype):
    if op is Op.AVG:
        # gloo has no AVG; emulate with SUM + divide
        return None
    if op not in GLOO_OP_MAP:
        raise ValueError(f"unsupported reduction {op} on the CPU backend")
    return GLOO_OP_MAP[op]


def allreduce(x, op, comm):
    _check_bitwise_dtype(op, x, "allreduce")
    with debug_timer("Allreduce", comm.rank, f"{x.numel()} items"):
        out = x.clone().contiguous()
        if comm.size == 1:
            if op is Op.AVG:
                return out
            return out
        # gloo refuses bitwise ops on bool tensors; ride a uint8 view
        # (bitwise on 0/1 == logical)
        as_bool = x.dtype == torch.bool and op in _BITWISE_OPS
        if as_bool:
            out = out.to(torch.uint8)
        gop = _gloo_reduce_op(op, x.dtype)
        if gop is None:  # AVG
            dist.all_reduce(out, op=dist.ReduceOp.SUM, group=comm.gloo_group)
            out = out / comm.size
        else:
            dist.all_reduce(out, op=gop, group=comm.gloo_group)
        return out.to(torch.bool) if as_bool else out


def reduce(x, op, root, comm):
    _check_bitwise_dtype(op, x, "reduce")
    with debug_timer("Reduce", comm.rank, f"{x.numel()} items"):
        out = x.clone().contiguous()
        if comm.size == 1:
            return out if comm.rank == root else None
        as_bool = x.dtype == torch.bool and op in _BITWISE_OPS
        if as_bool:
            out = out.to(torch.uint8)
        gop = _gloo_reduce_op(op, x.dtype)
        if gop is None:
            dist.reduce(
                out, dst=comm.global_rank(root), op=dist.ReduceOp.SUM,
                group=comm.gloo_group,
            )
            if comm.rank == root:
                out = out / comm.size
        else:
            dist.reduce(
                out, dst=comm.global_rank(root), op=gop, group=comm.gloo_group
            )
        if comm.rank != root:
            return None
        return out.to(torch.bool) if as_bool else out


def allgather(x, comm):
    with debug_timer("Allgather", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        if comm.size == 1:
            return xc[None].clone()
        if xc.numel() == 0:
            return torch.empty((comm.size,) + tuple(x.shape), dtype=x.dtype)
        nbytes = xc.numel() * xc.element_size()
        ob = torch.empty((comm.size, max(nbytes, 1)), dtype=torch.uint8)
        dist.all_gather(list(ob.unbind(0)), _wire(xc),
                        group=comm.gloo_group)
        return ob.flatten().view(x.dtype).reshape(
            (comm.size,) + tuple(x.shape))


def alltoall(x, comm):
    with debug_timer("Alltoall", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        if comm.size == 1:
            return xc.clone()
        out = torch.empty_like(xc)
        dist.all_to_all_single(_wire(out), _wire(xc), group=comm.gloo_group)
        return out


def barrier(comm):
    with debug_timer("Barrier", comm.rank):
        if comm.size > 1:
            dist.barrier(group=comm.gloo_group)


def bcast(x, root, comm):
    with debug_timer("Bcast", comm.rank, f"{x.numel()} items"):
        out = x.clone().contiguous()
        if comm.size > 1:
            dist.broadcast(_wire(out), src=comm.global_rank(root),
                           group=comm.gloo_group)
        return out


def gather(x, root, comm):
    with debug_timer("Gather", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        if comm.size == 1:
            return xc[None].clone() if comm.rank == root else None
        if xc.numel() == 0:
            return (torch.empty((comm.size,) + tuple(x.shape), dtype=x.dtype)
                    if comm.rank == root else None)
        if comm.rank == root:
            nbytes = xc.numel() * xc.element_size()
            ob = torch.empty((comm.size, max(nbytes, 1)), dtype=torch.uint8)
            dist.gather(
                _wire(xc), list(ob.unbind(0)), dst=comm.global_rank(root),
                group=comm.gloo_group,
            )
            return ob.flatten().view(x.dtype).reshape(
                (comm.size,) + tuple(x.shape))
        dist.gather(_wire(xc), None, dst=comm.global_rank(root),
                    group=comm.gloo_group)
        return None


def scatter(x, root, comm):
    with debug_timer("Scatter", comm.rank):
        if comm.size == 1:
            return x[0].clone().contiguous()
        if comm.rank == root:
            xc = x.contiguous()
            out = torch.empty(tuple(x.shape[1:]), dtype=x.dtype)
            dist.scatter(
                _wire(out), [_wire(t.contiguous()) for t in xc.unbind(0)],
                src=comm.global_rank(root), group=comm.gloo_group,
            )
        else:
            out = torch.empty(tuple(x.shape), dtype=x.dtype)
            dist.scatter(
                _wire(out), None, src=comm.global_rank(root),
                group=comm.gloo_group
            )
        return out


def scan(x, op, comm):
    """Inclusive prefix reduction: out_r = x_0 ⊕ ... ⊕ x_r.

    Implemented as a rank chain (recv partial from r-1, combine, forward to
    r+1) — the same dataflow the GPU ring uses with the HIP combine kernel.
    """
    _check_bitwise_dtype(op, x, "scan")
    with debug_timer("Scan", comm.rank, f"{x.numel()} items"):
        out = x.clone().contiguous()
        if comm.size == 1:
            return out
        if comm.rank > 0:
            partial = torch.empty_like(out)
            dist.recv(_wire(partial), src=comm.global_rank(comm.rank - 1),
                      group=comm.gloo_group, tag=771)
            out = combine(op, partial, out)
        if comm.rank < comm.size - 1:
            dist.send(_wire(out.contiguous()),
                      dst=comm.global_rank(comm.rank + 1),
                      group=comm.gloo_group, tag=771)
        return out


def _check_send_tag(tag, what):
    if tag == ANY_TAG:
        raise ValueError(f"{what}: ANY_TAG is not a sendable tag")


def send(x, dest, tag, comm):
    with debug_timer("Send", comm.rank, f"to {dest}, tag {tag}"):
        _check_send_tag(tag, "send")
        xc = x.contiguous().clone()
        if dest == comm.rank:
            _self_queue(comm).append((tag, xc))
            return
        envelope.box_for(comm).post(dest, tag,
                                    xc.numel() * xc.element_size())
        _isend_data(comm, _wire(xc), dest)


def recv(template, source, tag, comm, status):
    with debug_timer("Recv", comm.rank, f"from {source}, tag {tag}"):
        out = torch.empty(
            tuple(template.shape), dtype=template.dtype
        )
        if source == comm.rank:
            q = _self_queue(comm)
            for i, (t, buf) in enumerate(q):
                if tag in (ANY_TAG, t):
                    q.pop(i)
                    out.copy_(buf)
                    _fill_status(status, comm.rank, t, out)
                    return out
            raise RuntimeError(
                "recv from self with no matching buffered send"
            )
        if source == ANY_SOURCE and comm.size == 1:
            raise ValueError(
                "recv: ANY_SOURCE with no remote peers (comm size 1) — "
                "buffered self-sends need an explicit source=comm.rank"
            )
        return _env_recv_into(out, source, tag, comm, status)


def sendrecv(sendbuf, recvbuf, source, dest, sendtag, recvtag, comm, status):
    with debug_timer(
        "Sendrecv", comm.rank, f"src {source} dst {dest}"
    ):
        out = torch.empty(tuple(recvbuf.shape), dtype=recvbuf.dtype)
        if source == comm.rank and dest == comm.rank:
            out.copy_(sendbuf.reshape(out.shape))
            _fill_status(status, source, recvtag, out)
            return out
        sc = sendbuf.contiguous().clone()
        # same envelope plane as send/recv, so a sendrecv pairs with a
        # plain send or recv on the peer (the halo oracle does exactly
        # that); the data isend never blocks, so any issue order is
        # deadlock-free
        if dest == comm.rank:
            _self_queue(comm).append((sendtag, sc))
        else:
            _check_send_tag(sendtag, "sendrecv(sendtag)")
            envelope.box_for(comm).post(dest, sendtag,
                                        sc.numel() * sc.element_size())
            _isend_data(comm, _wire(sc), dest)
        if source == comm.rank:
            q = _self_queue(comm)
            for i, (t, buf) in enumerate(q):
                if recvtag in (ANY_TAG, t):
                    q.pop(i)
                    out.copy_(buf.reshape(out.shape))
                    _fill_status(status, comm.rank, t, out)
                    return out
            raise RuntimeError(
                "sendrecv from self with no matching buffered send"
            )
        return _env_recv_into(out, source, recvtag, comm, status)


def _fill_status(status, source, tag, out):
    if status is not None:
        status.source = source
        status.tag = tag
        status.count = out.numel() * out.element_size()
