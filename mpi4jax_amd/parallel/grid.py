"""Cartesian process grids and halo exchange.

The reference's only parallelism pattern is SPMD domain decomposition with
ordered halo exchanges (``/root/reference/examples/shallow_water.py:57-107``
for the grid, :172-264 for the clockwise exchange order that avoids
deadlock — SURVEY.md §2.4).  This module makes that pattern a first-class,
reusable component.

On the RCCL backend each direction's sendrecv is grouped, so a full
4-direction exchange is a handful of point-to-point xGMI transfers enqueued
on the compute stream with zero host synchronization.
"""

from .comm import resolve_comm


def default_dims(size):
    """Process grid (nproc_y, nproc_x) for a given world size.

    Mirrors the reference demo's choice (shallow_water.py:63-64):
    nproc_y = min(size, 2), nproc_x = size // nproc_y.
    """
    nproc_y = min(size, 2)
    if size % nproc_y != 0:
        raise ValueError(f"world size {size} not supported (must be even or 1)")
    return nproc_y, size // nproc_y


class CartesianGrid:
    """A 2-D process grid over a communicator.

    Rank r sits at coords ``(r // nx, r % nx)`` (row-major, matching
    ``np.unravel_index`` in the reference demo).
    """

    def __init__(self, comm=None, dims=None, periodic=(False, True)):
        self.comm = resolve_comm(comm)
        if dims is None:
            dims = default_dims(self.comm.size)
        self.nproc_y, self.nproc_x = dims
        if self.nproc_y * self.nproc_x != self.comm.size:
            raise ValueError(
                f"grid {dims} does not match comm size {self.comm.size}"
            )
        self.periodic_y, self.periodic_x = periodic
        self.coords = (
            self.comm.rank // self.nproc_x,
            self.comm.rank % self.nproc_x,
        )

    def rank_at(self, iy, ix):
        return iy * self.nproc_x + ix

    def neighbor(self, direction):
        """Rank of the neighbor in a direction, or None at a closed edge.

        Directions follow the reference demo's convention
        (shallow_water.py:210-224): y grows northward, x grows eastward.
        """
        iy, ix = self.coords
        if direction == "south":
            iy2, ix2 = iy - 1, ix
        elif direction == "north":
            iy2, ix2 = iy + 1, ix
        elif direction == "west":
            iy2, ix2 = iy, ix - 1
        elif direction == "east":
            iy2, ix2 = iy, ix + 1
        else:
            raise ValueError(f"bad direction {direction}")
        if not 0 <= iy2 < self.nproc_y:
            if not self.periodic_y:
                return None
            iy2 %= self.nproc_y
        if not 0 <= ix2 < self.nproc_x:
            if not self.periodic_x:
                return None
            ix2 %= self.nproc_x
        return self.rank_at(iy2, ix2)

    # halo slice tables for a (ny, nx) array with a 1-cell halo ring.
    # The reference demo exchanges in clockwise W,N,E,S order
    # (shallow_water.py:180-208); we use both column exchanges first, then
    # both row exchanges — full-row sends then carry fresh column-halo
    # corners, which satisfies every corner dependency the clockwise order
    # satisfies (and makes the SW corner fresh too), while letting the
    # fused GPU path batch each half into a single RCCL group.
    _SEND_ROW = {"south": 1, "north": -2}
    _RECV_ROW = {"south": 0, "north": -1}
    _SEND_COL = {"west": 1, "east": -2}
    _RECV_COL = {"west": 0, "east": -1}
    _ORDER = (("west", "east"), ("east", "west"),
              ("north", "south"), ("south", "north"))

    def _get_edge(self, arr, direction, kind):
        if direction in ("south", "north"):
            row = (self._SEND_ROW if kind == "send" else self._RECV_ROW)[direction]
            return arr[row, :]
        col = (self._SEND_COL if kind == "send" else self._RECV_COL)[direction]
        return arr[:, col]

    def _set_edge(self, arr, direction, value):
        if direction in ("south", "north"):
            arr[self._RECV_ROW[direction], :] = value
        else:
            arr[:, self._RECV_COL[direction]] = value

    def halo_exchange(self, arr):
        """Exchange the 1-cell halo ring of a local 2-D array.

        Returns a NEW tensor (the input is never mutated — the reference's
        immutability contract).  Closed-edge halos are left as they were.
        """
        return self._halo_exchange_impl(arr.clone())

    def halo_exchange_(self, arr):
        """In-place halo exchange (internal fast path for owned buffers)."""
        return self._halo_exchange_impl(arr)

    def _halo_exchange_impl(self, out):
        # imported here to avoid a circular import at package init
        from ..ops.sendrecv import sendrecv
        from ..ops.send import send
        from ..ops.recv import recv

        me = self.comm.rank
        for send_dir, recv_dir in self._ORDER:
            send_to = self.neighbor(send_dir)
            recv_from = self.neighbor(recv_dir)
            if send_to is None and recv_from is None:
                continue
            if send_to == me and recv_from == me:
                # periodic self-wrap: a plain on-device copy
                self._set_edge(out, recv_dir,
                               self._get_edge(out, send_dir, "send"))
                continue
            if send_to is None:
                got = recv(self._get_edge(out, recv_dir, "recv"),
                           source=recv_from, comm=self.comm)
                self._set_edge(out, recv_dir, got)
            elif recv_from is None:
                send(self._get_edge(out, send_dir, "send"), dest=send_to,
                     comm=self.comm)
            else:
                got = sendrecv(
                    self._get_edge(out, send_dir, "send"),
                    self._get_edge(out, recv_dir, "recv"),
                    source=recv_from,
                    dest=send_to,
                    comm=self.comm,
                )
                self._set_edge(out, recv_dir, got)
        return out
