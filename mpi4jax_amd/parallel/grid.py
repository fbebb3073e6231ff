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


def halo_plan(nx, ny):
    """The halo-exchange transfer plan for a (ny, nx) array.

    The reference demo exchanges full edges in clockwise W,N,E,S order so
    later phases carry corner values (shallow_water.py:180-208).  This plan
    is bit-identical but fully order-independent: full columns, *interior*
    rows, and the four halo corners exchanged directly with the diagonal
    neighbors (a corner equals the diagonal neighbor's interior corner
    cell).  Order-independence is what lets the fused GPU path issue the
    entire exchange as ONE RCCL group.

    Entries:
      ("cols",   send_dir, recv_dir, recv_col, send_col)
      ("rows",   send_dir, recv_dir, recv_row, send_row)   # cols 1..nx-2
      ("corner", send_diag, recv_diag, recv_cell, send_cell)
    The only ordering requirement: column writes land before corner writes
    (both touch the 4 corner cells; corners win) — the executors below and
    in the fused path preserve that.
    """
    return (
        ("cols", "west", "east", nx - 1, 1),
        ("cols", "east", "west", 0, nx - 2),
        ("rows", "north", "south", 0, ny - 2),
        ("rows", "south", "north", ny - 1, 1),
        # recv corner (rj, ri) <- opposite diagonal's interior corner
        ("corner", (-1, -1), (1, 1), (ny - 1, nx - 1), (1, 1)),
        ("corner", (-1, 1), (1, -1), (ny - 1, 0), (1, nx - 2)),
        ("corner", (1, -1), (-1, 1), (0, nx - 1), (ny - 2, 1)),
        ("corner", (1, 1), (-1, -1), (0, 0), (ny - 2, nx - 2)),
    )


def halo_exchange_schedule(grid, nx, ny):
    """Resolve halo_plan against a grid into concrete transfer ops.

    Returns ``(wrap_sides, col_ops, row_ops, cor_ops, cor_mask)``:
      wrap_sides: kernel-side periodic wraps (0=east halo, 1=west, 2=both)
      col_ops:  [(k, send_to, recv_from, send_col, recv_col)]
      row_ops:  [(send_to, recv_from, recv_row, send_row)]  (interior cols)
      cor_ops:  [(d, send_to, recv_from)]  (diagonal corner transfers)
    All peer ranks are comm-relative; every transfer list entry has at
    least one non-None peer.  The cross-rank matching of this schedule
    (send sequences == recv sequences per ordered rank pair) is verified
    for every topology in tests/test_shallow_water.py.
    """
    me = grid.comm.rank
    plan = halo_plan(nx, ny)
    wrap_sides, col_ops, row_ops, cor_ops = [], [], [], []
    col_nbrs = [(grid.neighbor(sdir), grid.neighbor(rdir))
                for _, sdir, rdir, _, _ in plan[:2]]
    if all(st == me and rf == me for st, rf in col_nbrs):
        wrap_sides.append(2)  # both periodic wraps in one kernel
        col_nbrs = [(None, None), (None, None)]
    for k, (_, sdir, rdir, ridx, sidx) in enumerate(plan[:2]):
        st, rf = col_nbrs[k]
        if st is None and rf is None:
            continue
        if st == me and rf == me:
            wrap_sides.append(0 if ridx == nx - 1 else 1)
            continue
        col_ops.append((k, st, rf, sidx, ridx))
    for (_, sdir, rdir, ridx, sidx) in plan[2:4]:
        st, rf = grid.neighbor(sdir), grid.neighbor(rdir)
        if st is not None or rf is not None:
            row_ops.append((st, rf, ridx, sidx))
    cor_mask = 0
    for d, (_, sdiag, rdiag, _, _) in enumerate(plan[4:]):
        st = grid.neighbor2(*sdiag)
        rf = grid.neighbor2(*rdiag)
        if st is not None or rf is not None:
            cor_ops.append((d, st, rf))
            if rf is not None:
                cor_mask |= 1 << d
    return wrap_sides, col_ops, row_ops, cor_ops, cor_mask


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

    _DIRS = {"south": (-1, 0), "north": (1, 0), "west": (0, -1),
             "east": (0, 1)}

    def neighbor(self, direction):
        """Rank of the neighbor in a direction, or None at a closed edge.

        Directions follow the reference demo's convention
        (shallow_water.py:210-224): y grows northward, x grows eastward.
        """
        dy, dx = self._DIRS[direction]
        return self.neighbor2(dy, dx)

    def neighbor2(self, dy, dx):
        """Rank at a (possibly diagonal) grid offset, or None when closed."""
        iy2 = self.coords[0] + dy
        ix2 = self.coords[1] + dx
        if not 0 <= iy2 < self.nproc_y:
            if not self.periodic_y:
                return None
            iy2 %= self.nproc_y
        if not 0 <= ix2 < self.nproc_x:
            if not self.periodic_x:
                return None
            ix2 %= self.nproc_x
        return self.rank_at(iy2, ix2)

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
        if out.dim() != 2 or out.shape[0] < 3 or out.shape[1] < 3:
            raise ValueError(
                f"halo_exchange needs a 2-D array of at least 3x3 "
                f"(1-cell halo ring + interior), got {tuple(out.shape)}"
            )
        ny, nx = out.shape

        def xchg(send_to, recv_from, send_view, recv_setter, template):
            if send_to is None and recv_from is None:
                return
            if send_to is None:
                recv_setter(recv(template, source=recv_from, comm=self.comm))
            elif recv_from is None:
                send(send_view, dest=send_to, comm=self.comm)
            else:
                recv_setter(sendrecv(send_view, template, source=recv_from,
                                     dest=send_to, comm=self.comm))

        for kind, *spec in halo_plan(nx, ny):
            if kind == "cols":
                sdir, rdir, ridx, sidx = spec
                st, rf = self.neighbor(sdir), self.neighbor(rdir)
                if st == me and rf == me:
                    out[:, ridx] = out[:, sidx]  # periodic self-wrap copy
                    continue

                def set_col(v, ridx=ridx):
                    out[:, ridx] = v

                xchg(st, rf, out[:, sidx], set_col, out[:, ridx])
            elif kind == "rows":
                sdir, rdir, ridx, sidx = spec
                st, rf = self.neighbor(sdir), self.neighbor(rdir)

                def set_row(v, ridx=ridx):
                    out[ridx, 1:-1] = v

                xchg(st, rf, out[sidx, 1:-1].contiguous(), set_row,
                     out[ridx, 1:-1])
            else:  # corner
                (sdy, sdx), (rdy, rdx), (rj, ri), (sj, si) = spec
                st, rf = self.neighbor2(sdy, sdx), self.neighbor2(rdy, rdx)

                def set_corner(v, rj=rj, ri=ri):
                    out[rj, ri] = v[0]

                xchg(st, rf, out[sj, si:si + 1].contiguous(), set_corner,
                     out[rj, ri:ri + 1])
        return out
