"""Communicator: process bootstrap + RCCL communicator lifecycle.

Reference model: one process per device; every op takes
``comm=None`` which defaults to a lazily-created *clone* of the world
communicator, isolating library traffic from user traffic
(``/root/reference/mpi4jax/_src/utils.py:20-27``,
``docs/sharp-bits.rst:74-135``).

MI355X-native equivalent:

* process model is 1 process per GPU (torchrun / env:// rendezvous over
  127.0.0.1);
* the *bootstrap* plane is a gloo process group (tiny CPU traffic only:
  rendezvous + 128-byte ncclUniqueId exchange + CPU-tensor collectives for
  the GPU-less test path);
* the *data* plane for device tensors is an ``ncclComm_t`` per communicator,
  owned by the native extension and driven entirely on the current HIP
  stream (zero sync, zero staging) — see ``csrc/bridge.cpp``.

``Clone()`` creates a fresh gloo group *and* a fresh RCCL communicator, so a
clone's traffic can never be mismatched with its parent's — the same
isolation rationale as the reference's ``Comm.Clone()`` semantics.
"""

import os
import threading

import torch
import torch.distributed as dist

_LOCK = threading.RLock()
_WORLD = None
_DEFAULT_COMM = None
_COMM_COUNTER = [0]


def _env_int(name, default):
    v = os.environ.get(name)
    if v is None or v == "":
        return default
    return int(v)


class Communicator:
    """A communication context over a fixed set of ranks.

    Mirrors the mpi4py surface the reference uses: ``Get_rank``,
    ``Get_size``, ``Clone``, ``Split``.
    """

    def __init__(self, *, ranks, gloo_group, parent_world_rank, label):
        # global (world) ranks participating, sorted by comm rank
        self._ranks = list(ranks)
        self._gloo_group = gloo_group
        self._world_rank = parent_world_rank
        self._label = label
        self._rank = self._ranks.index(parent_world_rank)
        self._size = len(self._ranks)
        self._rccl_id = None  # int64 key into the native comm registry
        self._freed = False

    # -- mpi4py-compatible surface -------------------------------------
    def Get_rank(self):
        return self._rank

    def Get_size(self):
        return self._size

    @property
    def rank(self):
        return self._rank

    @property
    def size(self):
        return self._size

    def __repr__(self):
        return (
            f"Communicator({self._label}, rank={self._rank}, "
            f"size={self._size})"
        )

    # -- lifecycle ------------------------------------------------------
    def Clone(self):
        """New communicator over the same ranks with isolated traffic."""
        return _new_comm(self._ranks, f"{self._label}.clone")

    def Split(self, color, key=0):
        """Split into sub-communicators by color, ordered by (key, rank).

        Collective over this communicator (group creation uses local
        synchronization, so only this communicator's members take part —
        splitting a sub-communicator works).
        """
        info = self._allgather_py((int(color), int(key)))
        groups = {}
        for comm_rank, (c, k) in enumerate(info):
            groups.setdefault(c, []).append((k, comm_rank))
        members = [self._ranks[r] for _, r in sorted(groups[int(color)])]
        return _new_comm(members, f"{self._label}.split{int(color)}")

    def free(self):
        """Release the native RCCL communicator (gloo groups are pooled)."""
        if self._rccl_id is not None:
            from .._backend import rccl

            rccl.ext().comm_destroy(self._rccl_id)
            self._rccl_id = None
        self._freed = True

    # -- internals ------------------------------------------------------
    @property
    def gloo_group(self):
        return self._gloo_group

    def global_rank(self, comm_rank):
        """Translate a comm-relative rank to the torch.distributed rank."""
        if not 0 <= comm_rank < self._size:
            raise ValueError(
                f"invalid rank {comm_rank} for communicator of size "
                f"{self._size}"
            )
        return self._ranks[comm_rank]

    def _allgather_py(self, obj):
        """Allgather a small picklable object over the bootstrap plane."""
        if self._size == 1:
            return [obj]
        out = [None] * self._size
        dist.all_gather_object(out, obj, group=self._gloo_group)
        return out

    def rccl_handle(self):
        """Lazily create the native RCCL communicator for this comm.

        The 128-byte ncclUniqueId is generated on comm rank 0 and exchanged
        over the bootstrap plane; ``ncclCommInitRank`` then runs in the
        native extension.  Collective on first GPU use of this comm.
        """
        if self._rccl_id is not None:
            return self._rccl_id
        with _LOCK:
            if self._rccl_id is not None:
                return self._rccl_id
            from .._backend import rccl

            ext = rccl.ext()
            if self._rank == 0:
                uid = ext.get_unique_id()  # bytes
            else:
                uid = None
            if self._size > 1:
                uid_t = torch.empty(128, dtype=torch.uint8)
                if self._rank == 0:
                    uid_t.copy_(torch.frombuffer(bytearray(uid), dtype=torch.uint8))
                dist.broadcast(
                    uid_t, src=self.global_rank(0), group=self._gloo_group
                )
                uid = bytes(uid_t.tolist())
            self._rccl_id = ext.comm_init_rank(self._size, self._rank, uid)
            return self._rccl_id


def init(device=None):
    """Initialize the world communicator.

    Reads the torchrun/env:// variables (RANK, WORLD_SIZE, MASTER_ADDR,
    MASTER_PORT, LOCAL_RANK).  Single-process with no env works without any
    rendezvous.  Idempotent.  Importing the package does NOT initialize
    anything (unlike the reference, whose import runs MPI_Init,
    ``_src/__init__.py:1-3``) — but the first collective call does, so
    user-visible behavior matches.
    """
    global _WORLD
    if _WORLD is not None:  # lock-free fast path (also torch.compile-safe)
        return _WORLD
    with _LOCK:
        if _WORLD is not None:
            return _WORLD
        world_size = _env_int("WORLD_SIZE", 1)
        if world_size > 1 and not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group("gloo")
        if dist.is_initialized():
            world_size = dist.get_world_size()
            rank = dist.get_rank()
            group = dist.group.WORLD
        else:
            rank = 0
            world_size = 1
            group = None
        # one process per GPU
        if torch.cuda.is_available():
            local_rank = _env_int("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1))
            if device is None:
                device = local_rank % torch.cuda.device_count()
            torch.cuda.set_device(device)
        _WORLD = Communicator(
            ranks=list(range(world_size)),
            gloo_group=group,
            parent_world_rank=rank,
            label="WORLD",
        )
        return _WORLD


def _new_comm(ranks, label):
    """Create a communicator over the given global ranks.

    Collective over ``ranks`` only (``use_local_synchronization``), so
    sub-communicators can be cloned/split without involving the world.
    """
    world = init()
    if world.size == 1:
        _COMM_COUNTER[0] += 1
        return Communicator(
            ranks=[0], gloo_group=None, parent_world_rank=0,
            label=f"{label}#{_COMM_COUNTER[0]}",
        )
    group = dist.new_group(ranks, use_local_synchronization=True)
    _COMM_COUNTER[0] += 1
    return Communicator(
        ranks=ranks,
        gloo_group=group,
        parent_world_rank=world._world_rank,
        label=f"{label}#{_COMM_COUNTER[0]}",
    )


def get_world():
    return init()


def get_default_comm():
    """Default communicator: a lazily-created clone of the world.

    Mirrors ``/root/reference/mpi4jax/_src/utils.py:20-27`` — cloning
    isolates this library's traffic from any other torch.distributed use.
    """
    global _DEFAULT_COMM
    if _DEFAULT_COMM is not None:  # lock-free fast path (compile-safe)
        return _DEFAULT_COMM
    with _LOCK:
        if _DEFAULT_COMM is None:
            _DEFAULT_COMM = init().Clone()
        return _DEFAULT_COMM


class _WorldProxy:
    """Lazy stand-in so ``mpi4jax_amd.COMM_WORLD`` can be passed as comm=."""

    def _resolve(self):
        return get_world()

    def __getattr__(self, name):
        return getattr(self._resolve(), name)

    def __repr__(self):
        return "COMM_WORLD"


COMM_WORLD = _WorldProxy()


def resolve_comm(comm):
    if comm is None:
        return get_default_comm()
    if isinstance(comm, _WorldProxy):
        return get_world()
    if isinstance(comm, Communicator):
        return comm
    raise TypeError(
        f"comm must be a mpi4jax_amd.Communicator or None, got {type(comm)}"
    )


def flush():
    """Drain pending device-side communication (atexit hook).

    Analog of the reference's atexit ``jax.effects_barrier()``
    (``_src/__init__.py:14-24``): everything enqueued is stream-ordered, so
    a device synchronize makes it all visible before exit.
    """
    try:
        if torch.cuda.is_available() and torch.cuda.is_initialized():
            torch.cuda.synchronize()
        from .._backend import cpu as _cpu

        _cpu.drain_pending()  # MPI_Finalize semantics for in-flight sends
        from .._backend import rccl

        if rccl.ext_is_loaded():
            rccl.ext().check_async_errors()
    except ImportError:
        pass
    except Exception:
        # an async comm failure at exit must never vanish silently
        import sys
        import traceback

        print("mpi4jax_amd.flush(): pending communication failed:",
              file=sys.stderr)
        traceback.print_exc()
        raise


def finalize():
    """Destroy all native communicators and the process group."""
    global _WORLD, _DEFAULT_COMM
    flush()
    with _LOCK:
        try:
            from .._backend import rccl

            if rccl.ext_is_loaded():
                rccl.ext().destroy_all_comms()
        except Exception:
            pass
        # destroy_all_comms freed every native handle: drop the stale ids
        # so a later re-init() creates fresh ones instead of dangling, and
        # clear the jit_ops registry (cached compiled graphs re-resolve
        # the default key at next call because _DEFAULT_COMM resets too).
        from ..ops import jit_ops

        for c in list(jit_ops._COMMS.values()):
            c._rccl_id = None
        jit_ops._COMMS.clear()
        jit_ops._KEYS.clear()
        if _WORLD is not None:
            _WORLD._rccl_id = None
        if _DEFAULT_COMM is not None:
            _DEFAULT_COMM._rccl_id = None
        _WORLD = None
        _DEFAULT_COMM = None
        if dist.is_initialized():
            dist.destroy_process_group()
