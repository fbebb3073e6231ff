"""GPU transport: the hand-written HIP/RCCL native extension.

Replaces the reference's CUDA FFI backend
(``/root/reference/mpi4jax/_src/xla_bridge/mpi_xla_bridge_cuda.cpp``) with a
true zero-copy, zero-sync design: every collective is enqueued by
``csrc/bridge.cpp`` on the *current* HIP stream via RCCL, operating directly
on the tensor's HBM3E buffer.  There is no ``hipStreamSynchronize`` and no
host staging anywhere on this path (the reference's COPY_TO_HOST mode,
``mpi_xla_bridge_cuda.cpp:185-201``, simply does not exist).

On a GPU box this module refuses to fall back: if the extension is missing,
every op raises ImportError loudly rather than silently using eager torch.

Dtype strategy:
* pure data movement (allgather/alltoall/bcast/gather/scatter/p2p) ships
  raw bytes — every torch dtype travels;
* arithmetic reductions use the native RCCL dtype; complex reduces as
  pairs of reals (SUM/AVG only), bool maps to uint8 with OR/AND semantics,
  int16 upcasts to int32 on-stream (RCCL has no int16).
"""

import weakref

import torch

from ..ops.reduce_ops import Op, RCCL_OP_ENUM, BITWISE_OP_ENUM
from ..utils.status import ANY_SOURCE, ANY_TAG
from ..utils.logging import debug_timer, get_logging

_EXT = None
_EXT_ERR = None

# per-comm FIFO for unmatched self-sends: an ncclSend to self outside a
# group would block the stream forever, so self p2p short-circuits to a
# stream-ordered device copy (MPI buffered-self-send semantics).
# Weak-keyed so a GC'd communicator can never alias a new one's queue.
_SELF_QUEUES = weakref.WeakKeyDictionary()


def ext():
    """Import the native extension (built in-tree by setup.py/__graft_entry__)."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    if _EXT_ERR is not None:
        raise _EXT_ERR
    try:
        import os

        if os.environ.get("MPI4JAX_AMD_SW_EXT", "") == "nofma":
            # FMA-contraction control build: device kernels compiled with
            # -ffp-contract=off (tests/test_gpu_nofma.py)
            from .. import _rccl_C_nofma as _rccl_C  # type: ignore
        else:
            from .. import _rccl_C  # type: ignore

        _check_build_compat(_rccl_C)
        _EXT = _rccl_C
        _EXT.set_logging(get_logging())
        # opt-in fail-fast watchdog: a wedged p2p (mismatched send/recv,
        # dead peer) aborts all communicators and exits loudly after this
        # many seconds instead of hanging the stream forever
        wd = os.environ.get("MPI4JAX_AMD_WATCHDOG_SEC", "").strip()
        if wd:
            _EXT.set_watchdog(float(wd))
        return _EXT
    except ImportError as e:  # pragma: no cover - GPU box only
        _EXT_ERR = ImportError(
            "mpi4jax_amd native HIP/RCCL extension (_rccl_C) is not built. "
            "GPU collectives REQUIRE it — there is no eager fallback. "
            "Build it with `python setup.py build_ext --inplace` "
            f"(original error: {e})"
        )
        raise _EXT_ERR


def _check_build_compat(ext_mod):
    """Runtime-vs-build compatibility guard (the reference's MPI ABI
    check, ``xla_bridge/__init__.py:23-89``, re-expressed for the pair
    that matters here: the torch C++ ABI the extension was compiled
    against, and the RCCL header vs the librccl actually loaded).

    torch major.minor mismatch is an error (the C++ ABI is not stable
    across minor releases; symptoms otherwise range from import errors
    to silent corruption); RCCL header-vs-runtime skew is a warning
    (the NCCL API is stable within a major version).  Bypass with
    ``MPI4JAX_AMD_SKIP_ABI_CHECK=1`` — same escape hatch semantics as
    the reference's ``MPI4JAX_SKIP_ABI_CHECK``.
    """
    import os
    import warnings

    if os.environ.get("MPI4JAX_AMD_SKIP_ABI_CHECK", "") == "1":
        return
    try:
        built = dict(ext_mod.build_info())
    except AttributeError:  # older .so without build_info
        return
    run_torch = torch.__version__.split("+")[0]
    b = built["torch"].split(".")[:2]
    r = run_torch.split(".")[:2]
    if b != r:
        raise ImportError(
            f"mpi4jax_amd native extension was built against torch "
            f"{built['torch']} but torch {run_torch} is running — the "
            "torch C++ ABI is not stable across versions; rebuild with "
            "`python setup.py build_ext --inplace` (or set "
            "MPI4JAX_AMD_SKIP_ABI_CHECK=1 to proceed at your own risk)"
        )
    rt = dict(ext_mod.version_info()).get("rccl", 0)
    bh = built.get("rccl_header", 0)
    if rt and bh and rt // 10000 != bh // 10000:
        raise ImportError(
            f"RCCL major-version mismatch: extension built against "
            f"header {bh} but librccl {rt} is loaded (set "
            "MPI4JAX_AMD_SKIP_ABI_CHECK=1 to override)"
        )
    if rt and bh and rt != bh:
        warnings.warn(
            f"mpi4jax_amd: RCCL header/runtime skew (built {bh}, "
            f"running {rt}) — API-compatible within a major version, "
            "continuing",
            stacklevel=3,
        )


def ext_is_loaded():
    return _EXT is not None


# comm -> the HIP stream the previous collective was enqueued on.  The
# reference's JAX ordered effect (utils.py:45-53 token threading) pinned
# "program order == network order" against compiler reordering; the torch
# analog of that hazard is a user enqueuing collectives on DIFFERENT
# streams: per-stream they'd execute in submission race order, and a
# reorder against another rank's sequence deadlocks RCCL.  The fence makes
# each collective's stream wait on the previous collective's stream
# (event bridge), so per-communicator program order == network order even
# across streams — the ordered-effect guarantee, enforced in hardware.
_ORDER_STREAMS = weakref.WeakKeyDictionary()


def _order_fence(comm):
    if torch.cuda.is_current_stream_capturing():
        # an event bridge from a non-capture stream is illegal inside
        # capture; the graph path pins its own stream discipline
        # (models/shallow_water.make_stepper captures on ONE stream)
        return
    cur = torch.cuda.current_stream()
    prev = _ORDER_STREAMS.get(comm)
    if prev is not None and prev != cur:
        ev = torch.cuda.Event()
        ev.record(prev)
        cur.wait_event(ev)
    _ORDER_STREAMS[comm] = cur


def _handle(comm):
    _order_fence(comm)
    return comm.rccl_handle()


def _bytes(x):
    """Byte view of a contiguous tensor (shares storage)."""
    return x.flatten().view(torch.uint8)


def _check_op(op, x, op_name):
    if op in BITWISE_OP_ENUM:
        # bitwise rides the HIP-combine p2p compositions (scan ring /
        # scan+bcast), valid for integer and bool dtypes only
        if x.is_floating_point() or x.is_complex():
            raise ValueError(
                f"{op_name}: {op} requires an integer or bool dtype, "
                f"got {x.dtype}"
            )
        return
    if op not in RCCL_OP_ENUM:
        raise ValueError(
            f"{op_name}: reduction {op} is not supported on the RCCL "
            f"backend (supported: SUM, PROD, MIN, MAX, AVG, and "
            f"BAND/BOR/BXOR on integer dtypes)"
        )
    if x.is_complex() and op not in (Op.SUM, Op.AVG):
        raise ValueError(f"{op_name}: {op} is undefined for complex dtypes")


# bool reductions: OR for SUM/MAX, AND for PROD/MIN (MPI logical semantics)
_BOOL_OP = {Op.SUM: Op.MAX, Op.MAX: Op.MAX, Op.PROD: Op.MIN, Op.MIN: Op.MIN}


def _reduction_view(x, op, op_name):
    """Map (tensor, op) onto an RCCL-native (tensor, op, postprocess)."""
    if x.is_complex():
        return torch.view_as_real(x), op, None
    if x.dtype == torch.bool:
        if op in BITWISE_OP_ENUM:  # bitwise on bool == logical, via uint8
            return x.to(torch.uint8), op, lambda t: t.to(torch.bool)
        if op not in _BOOL_OP:
            raise ValueError(f"{op_name}: {op} undefined for bool")
        return x.to(torch.uint8), _BOOL_OP[op], lambda t: t.to(torch.bool)
    if x.dtype == torch.int16:
        return x.to(torch.int32), op, lambda t: t.to(torch.int16)
    return x, op, None


def allreduce(x, op, comm):
    _check_op(op, x, "allreduce")
    if x.numel() == 0:
        return x.clone()
    if op in BITWISE_OP_ENUM:
        # RCCL has no bitwise reductions: inclusive scan (ring + HIP
        # combine kernel) leaves the full reduction on the last rank,
        # then broadcast it.  O(P) latency; exact (ops are associative
        # and commutative, order is deterministic).
        full = scan(x, op, comm)
        return bcast(full, comm.size - 1, comm)
    with debug_timer("Allreduce", comm.rank, f"{x.numel()} items"):
        xr, rop, post = _reduction_view(x.contiguous(), op, "allreduce")
        out = torch.empty_like(xr)
        ext().allreduce(out, xr, RCCL_OP_ENUM[rop], _handle(comm))
        if post is not None:
            out = post(out)
        if x.is_complex():
            out = torch.view_as_complex(out)
        return out.reshape(x.shape)


def reduce(x, op, root, comm):
    _check_op(op, x, "reduce")
    if op in BITWISE_OP_ENUM:
        out = scan(x, op, comm)
        out = bcast(out, comm.size - 1, comm)  # see allreduce
        return out if comm.rank == root else None
    with debug_timer("Reduce", comm.rank, f"{x.numel()} items"):
        xr, rop, post = _reduction_view(x.contiguous(), op, "reduce")
        out = torch.empty_like(xr)
        ext().reduce(out, xr, RCCL_OP_ENUM[rop], root, _handle(comm))
        if comm.rank != root:
            return None
        if post is not None:
            out = post(out)
        if x.is_complex():
            out = torch.view_as_complex(out)
        return out.reshape(x.shape)


def scan(x, op, comm):
    _check_op(op, x, "scan")
    if x.numel() == 0:
        return x.clone()
    if op is Op.AVG:
        raise ValueError("scan: AVG is not a valid scan operator")
    with debug_timer("Scan", comm.rank, f"{x.numel()} items"):
        xr, rop, post = _reduction_view(x.contiguous(), op, "scan")
        out = torch.empty_like(xr)
        # the op code only ever reaches the HIP combine kernel in the
        # scan ring (bridge.cpp scan), so kernel-only bitwise codes are
        # legal here
        code = (BITWISE_OP_ENUM[rop] if rop in BITWISE_OP_ENUM
                else RCCL_OP_ENUM[rop])
        ext().scan(out, xr, code, _handle(comm))
        if post is not None:
            out = post(out)
        if x.is_complex():
            out = torch.view_as_complex(out)
        return out.reshape(x.shape)


def allgather(x, comm):
    with debug_timer("Allgather", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        out = torch.empty((comm.size,) + tuple(x.shape), dtype=x.dtype,
                          device=x.device)
        ext().allgather(_bytes(out), _bytes(xc), _handle(comm))
        return out


def alltoall(x, comm):
    with debug_timer("Alltoall", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        out = torch.empty_like(xc)
        ext().alltoall(_bytes(out), _bytes(xc), _handle(comm))
        return out


def barrier(comm):
    with debug_timer("Barrier", comm.rank):
        ext().barrier(_handle(comm))


def bcast(x, root, comm):
    with debug_timer("Bcast", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        out = torch.empty_like(xc)
        ext().broadcast(_bytes(out), _bytes(xc), root, _handle(comm))
        return out


def gather(x, root, comm):
    with debug_timer("Gather", comm.rank, f"{x.numel()} items"):
        xc = x.contiguous()
        if comm.rank == root:
            out = torch.empty((comm.size,) + tuple(x.shape), dtype=x.dtype,
                              device=x.device)
        else:
            out = torch.empty(0, dtype=x.dtype, device=x.device)
        ext().gather(_bytes(out), _bytes(xc), root, _handle(comm))
        return out if comm.rank == root else None


def scatter(x, root, comm):
    with debug_timer("Scatter", comm.rank):
        xc = x.contiguous()
        if comm.rank == root:
            out = torch.empty(tuple(x.shape[1:]), dtype=x.dtype,
                              device=x.device)
        else:
            out = torch.empty(tuple(x.shape), dtype=x.dtype, device=x.device)
        ext().scatter(_bytes(out), _bytes(xc), root, _handle(comm))
        return out


def _self_queue(comm):
    return _SELF_QUEUES.setdefault(comm, [])


def _check_gpu_tag(tag, what):
    """RCCL has no message envelope: remote GPU p2p is matched by enqueue
    order, NOT by tag.  Two differently-tagged in-flight messages to one
    peer would mismatch silently — so any non-default tag on a remote GPU
    transfer fails loudly here (the reference transmits tags through MPI,
    mpi_ops_common.h:354-367; self-messages keep real tag matching via
    the local queue).  Export MPI4JAX_AMD_ALLOW_GPU_TAGS=1 to accept
    order-based matching knowingly.
    """
    import os

    if tag in (0, ANY_TAG):
        return
    if os.environ.get("MPI4JAX_AMD_ALLOW_GPU_TAGS", "") == "1":
        return
    raise ValueError(
        f"{what}: tag {tag} cannot be matched on the RCCL path (no message"
        " envelope — matching is by enqueue order). Use the default tag,"
        " or set MPI4JAX_AMD_ALLOW_GPU_TAGS=1 to accept order-based"
        " matching."
    )


def send(x, dest, tag, comm):
    with debug_timer("Send", comm.rank, f"to {dest}, tag {tag}"):
        xc = x.contiguous()
        if dest == comm.rank:
            _self_queue(comm).append((tag, xc.clone()))  # stream-ordered copy
            return
        from . import envelope

        if envelope.enabled() and comm.gloo_group is not None:
            if tag == ANY_TAG:
                raise ValueError("send: ANY_TAG is not a sendable tag")
            envelope.box_for(comm).post(dest, tag,
                                        xc.numel() * xc.element_size())
        else:
            _check_gpu_tag(tag, "send")
        ext().send(_bytes(xc), dest, _handle(comm))


def recv(template, source, tag, comm, status):
    from . import envelope

    env_on = envelope.enabled() and comm.gloo_group is not None
    if source == ANY_SOURCE and not env_on:
        raise ValueError(
            "recv: ANY_SOURCE on the RCCL backend requires the envelope "
            "plane (set MPI4JAX_AMD_GPU_ENVELOPE=1) — RCCL itself has no "
            "message envelope; alternatively pass an explicit source rank"
        )
    if source == ANY_SOURCE and comm.size == 1:
        raise ValueError(
            "recv: ANY_SOURCE with no remote peers (comm size 1) — "
            "buffered self-sends need an explicit source=comm.rank"
        )
    with debug_timer("Recv", comm.rank, f"from {source}, tag {tag}"):
        out = torch.empty(tuple(template.shape), dtype=template.dtype,
                          device=template.device)
        if source == comm.rank:
            q = _self_queue(comm)
            for i, (t, buf) in enumerate(q):
                if tag in (ANY_TAG, t):
                    q.pop(i)
                    out.copy_(buf.reshape(out.shape))
                    _fill_status(status, source, t, out)
                    return out
            raise RuntimeError("recv from self with no matching buffered send")
        if env_on:
            h = _handle(comm)

            def recv_bytes(src, nbytes):
                buf = torch.empty(nbytes, dtype=torch.uint8,
                                  device=out.device)
                ext().recv(buf, src, h)
                return buf

            s, t, data = envelope.box_for(comm).claim(source, tag,
                                                      recv_bytes)
            if data is not None:  # drained ahead of order into the stash
                out.copy_(data.view(out.dtype).reshape(out.shape))
            else:
                ext().recv(_bytes(out), s, h)
            _fill_status(status, s, t, out)
            return out
        _check_gpu_tag(tag, "recv")
        ext().recv(_bytes(out), source, _handle(comm))
        _fill_status(status, source, tag, out)
        return out


def sendrecv(sendbuf, recvbuf, source, dest, sendtag, recvtag, comm, status):
    if source == ANY_SOURCE:
        raise ValueError(
            "sendrecv: ANY_SOURCE is not supported on the RCCL backend "
            "(the grouped send+recv pair needs an explicit peer; use "
            "separate send/recv with MPI4JAX_AMD_GPU_ENVELOPE=1 for "
            "wildcard receives)"
        )
    with debug_timer("Sendrecv", comm.rank, f"src {source} dst {dest}"):
        out = torch.empty(tuple(recvbuf.shape), dtype=recvbuf.dtype,
                          device=recvbuf.device)
        sc = sendbuf.contiguous()
        if source == comm.rank and dest == comm.rank:
            out.copy_(sc.reshape(out.shape))  # stream-ordered device copy
        else:
            if dest != comm.rank:
                _check_gpu_tag(sendtag, "sendrecv(sendtag)")
            if source != comm.rank:
                _check_gpu_tag(recvtag, "sendrecv(recvtag)")
            ext().sendrecv(_bytes(sc), _bytes(out), source, dest,
                           _handle(comm))
        _fill_status(status, source, recvtag, out)
        return out


def _fill_status(status, source, tag, out):
    if status is not None:
        status.source = source
        status.tag = tag
        status.count = out.numel() * out.element_size()


def pack2d(view):
    """Gather a 2-D strided device view into a contiguous tensor using the
    LDS-staged CDNA4 pack kernel (MI355X-native replacement for
    ``Tensor.contiguous()`` on the halo path)."""
    out = torch.empty(view.shape, dtype=view.dtype, device=view.device)
    ext().pack2d(out, view)
    return out


def unpack2d(view, data):
    """Scatter contiguous ``data`` into a 2-D strided device view."""
    ext().unpack2d(view, data.contiguous())


class group:
    """Batch multiple p2p/collective enqueues into one RCCL group.

    MI355X-native extra (no reference equivalent): wraps
    ``ncclGroupStart``/``ncclGroupEnd`` so e.g. all four halo-exchange
    directions fuse into a single RCCL launch over xGMI.
    """

    def __enter__(self):
        ext().group_start()
        return self

    def __exit__(self, *exc):
        ext().group_end()
        return False
