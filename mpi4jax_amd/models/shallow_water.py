"""Distributed nonlinear shallow-water solver (torch, MI355X-native).

This is the framework's flagship demo/benchmark workload, matching the
reference demo's physics and decomposition so its published numbers are
directly comparable (``/root/reference/examples/shallow_water.py``; the
scheme is the public Sadourny C-grid solver from dionhaefner/shallow-water).
Domain decomposition is a 2-D :class:`CartesianGrid`; every model step does
~10 halo exchanges through mpi4jax_amd ``sendrecv``/``send``/``recv``
(reference call stack: SURVEY.md §3.4).

The reference's benchmark config (docs/shallow-water.rst:49-52) is a
(3600, 1800) float32 domain run for 0.1 model days; wall-clock per model
day is the headline metric.

Two step implementations share semantics: an eager torch-op path (CPU
and reference/debugging) and the fused MI355X path — by default ONE
LDS-tiled update+friction pass plus two boundary-ring kernels (stage
30/32; at world 1 the ring kernel also writes the periodic wrap refresh,
so a whole model step launches zero exchange kernels; multi-rank, a
two-field u'/v'-strip exchange runs between the ring kernels), with the
two-kernel pipeline (stage 19 tendencies+update, stage 27 friction,
selectable variants and scalar fallbacks) behind MPI4JAX_AMD_SW_NOFUSE,
a one-native-call halo exchange (merged pack launch, single RCCL group,
merged unpack launch), and validated hipGraph capture for multistep
replay at any world size.
"""

import math
import time
from collections import namedtuple

import torch

from ..parallel.comm import resolve_comm
from ..parallel.grid import CartesianGrid, halo_exchange_schedule

ModelState = namedtuple("ModelState", "h u v dh du dv")

DAY_IN_SECONDS = 86_400.0
GRAVITY = 9.81
DEPTH = 100.0
CORIOLIS_F = 2e-4
CORIOLIS_BETA = 2e-11
ADAMS_BASHFORTH_A = 1.5 + 0.1
ADAMS_BASHFORTH_B = -(0.5 + 0.1)

_I = slice(1, -1)  # interior
_L = slice(None, -2)  # shifted left/down
_R = slice(2, None)  # shifted right/up


class ShallowWater:
    """Nonlinear shallow-water model on a distributed C-grid.

    Arguments:
        nx, ny: *global interior* grid size (the demo default is 360×180,
            benchmark mode 3600×1800).
        comm: communicator (defaults to the world).
        dims: process grid (nproc_y, nproc_x); default mirrors the
            reference demo (min(size,2), size/nproc_y).
        device / dtype: tensor placement (float32 matches the reference).
        periodic_x: periodic east-west boundary (reference default True).
    """

    def __init__(self, nx=360, ny=180, dx=5e3, dy=5e3, *, comm=None,
                 dims=None, device="cpu", dtype=torch.float32,
                 periodic_x=True, lateral_viscosity=None, fused=None,
                 ref_friction_bug=False, _force_remote_exchange=False):
        self.comm = resolve_comm(comm)
        self.grid = CartesianGrid(self.comm, dims=dims,
                                  periodic=(False, periodic_x))
        self.device = torch.device(device)
        self.dtype = dtype
        self.dx, self.dy = float(dx), float(dy)
        self.periodic_x = periodic_x

        self.nx_global = nx + 2
        self.ny_global = ny + 2
        npy, npx = self.grid.nproc_y, self.grid.nproc_x
        if nx % npx or ny % npy:
            raise ValueError(
                f"domain {ny}x{nx} not divisible by process grid {npy}x{npx}"
            )
        self.nx_local = nx // npx + 2
        self.ny_local = ny // npy + 2

        iy, ix = self.grid.coords
        # global index of this rank's first (halo) row/col
        self.y0 = (self.ny_local - 2) * iy
        self.x0 = (self.nx_local - 2) * ix

        # coordinates (match the reference's layout: index -1..n along each
        # axis times the spacing, shallow_water.py:86-90)
        x_g = (torch.arange(-1, self.nx_global - 1, dtype=torch.float64)
               * self.dx)
        y_g = (torch.arange(-1, self.ny_global - 1, dtype=torch.float64)
               * self.dy)
        self.length_x = float(x_g[-2] - x_g[1])
        self.length_y = float(y_g[-2] - y_g[1])
        self.x_local = x_g[self.x0:self.x0 + self.nx_local]
        self.y_local = y_g[self.y0:self.y0 + self.ny_local]

        yy = self.y_local[:, None].expand(self.ny_local, self.nx_local)
        self.coriolis = (CORIOLIS_F + yy * CORIOLIS_BETA).to(
            self.device, self.dtype
        )
        if lateral_viscosity is None:
            lateral_viscosity = 1e-3 * CORIOLIS_F * self.dx ** 2
        self.lateral_viscosity = float(lateral_viscosity)

        # CFL time step (shallow_water.py:135)
        self.dt = 0.125 * min(self.dx, self.dy) / math.sqrt(GRAVITY * DEPTH)

        # ref_friction_bug: reproduce the reference demo's v-friction typo
        # (`nu*(v_shift - u)`, /root/reference/examples/shallow_water.py:
        # 386-391) for cross-framework trajectory validation — see
        # docs/PARITY.md.  Eager path only.
        self.ref_friction_bug = bool(ref_friction_bug)
        self._force_remote_exchange = bool(_force_remote_exchange)
        if self.ref_friction_bug and fused:
            raise ValueError(
                "ref_friction_bug=True requires the eager path (fused=False)"
            )

        # fused CDNA4 kernel path (GPU only): 2-3 stencil kernels + one
        # halo-exchange group per step instead of ~300 eager torch kernels
        if fused is None:
            fused = (self.device.type == "cuda"
                     and not self.ref_friction_bug)
        self.fused = bool(fused) and self.device.type == "cuda"
        self._fb = None  # fused buffers

    # ------------------------------------------------------------------
    def _at_north_edge(self):
        return self.grid.coords[0] == self.grid.nproc_y - 1

    def _at_east_edge(self):
        return self.grid.coords[1] == self.grid.nproc_x - 1

    def enforce_boundaries(self, arr, kind):
        """Halo exchange + physical wall conditions for one field."""
        arr = self.grid.halo_exchange(arr)
        if not self.periodic_x and kind == "u" and self._at_east_edge():
            arr[:, -2] = 0.0
        if kind == "v" and self._at_north_edge():
            arr[-2, :] = 0.0
        return arr

    # ------------------------------------------------------------------
    def initial_conditions(self):
        """Zonal jet in geostrophic balance + small perturbation."""
        dev, dt_ = self.device, self.dtype
        ny_g, nx_g = self.ny_global, self.nx_global

        # u depends only on y -> compute the global 1-D profile, then the
        # (cheap, rank-local) global cumsum for geostrophic h
        y_g = (torch.arange(-1, ny_g - 1, dtype=torch.float64) * self.dy)
        u_prof = 10.0 * torch.exp(
            -((y_g - 0.5 * self.length_y) ** 2) / (0.02 * self.length_x) ** 2
        )
        cor_prof = CORIOLIS_F + y_g * CORIOLIS_BETA
        h_geo = torch.cumsum(-self.dy * u_prof * cor_prof / GRAVITY, dim=0)
        h_geo = h_geo - h_geo.mean()

        x_l = self.x_local.to(dev)
        y_l = self.y_local.to(dev)
        u0 = (u_prof[self.y0:self.y0 + self.ny_local]
              .to(dev)[:, None].expand(self.ny_local, self.nx_local))
        pert = (
            0.2
            * torch.sin(x_l / self.length_x * 10 * math.pi)[None, :]
            * torch.cos(y_l / self.length_y * 8 * math.pi)[:, None]
        )
        h0 = (DEPTH
              + h_geo[self.y0:self.y0 + self.ny_local].to(dev)[:, None]
              + pert)
        h0 = h0.to(dt_).contiguous()
        u0 = u0.to(dt_).contiguous()
        v0 = torch.zeros_like(u0)

        h0 = self.enforce_boundaries(h0, "h")
        u0 = self.enforce_boundaries(u0, "u")
        v0 = self.enforce_boundaries(v0, "v")

        z = torch.zeros_like(h0)
        return ModelState(h0, u0, v0, z.clone(), z.clone(), z.clone())

    # ------------------------------------------------------------------
    def _edge_pad_interior(self, h):
        """hc = interior of h padded by edge replication (halo ring)."""
        hc = h.clone()
        hc[0, :] = hc[1, :]
        hc[-1, :] = hc[-2, :]
        hc[:, 0] = hc[:, 1]
        hc[:, -1] = hc[:, -2]
        return hc

    def step(self, state, first_step=False):
        """One model step (Euler on the first step, AB2 afterwards)."""
        if self.fused:
            return self._step_fused(state, first_step)
        return self._step_eager(state, first_step)

    # ------------------------------------------------------------------
    # fused GPU path: csrc/shallow_water.hip stencil kernels + in-place
    # halo exchange of h/u/v.  Produces the same solution as the eager path
    # (equal up to FMA contraction inside the fused kernels).
    def _fused_flags(self):
        g = self.grid
        return [
            int(g.neighbor("south") is not None),
            int(g.neighbor("north") is not None),
            int(g.neighbor("west") is not None),
            int(g.neighbor("east") is not None),
            int((not self.periodic_x) and self._at_east_edge()),
            int(self._at_north_edge()),
            # x halos are LOCAL periodic wraps, not remote exchanges
            int(self.periodic_x and g.nproc_x == 1
                and not self._force_remote_exchange),
        ]

    def _init_fused_buffers(self, state):
        def z():
            return torch.zeros(
                (self.ny_local, self.nx_local), dtype=self.dtype,
                device=self.device,
            )

        fb = {
            "h": state.h.detach().clone().contiguous(),
            "u": state.u.detach().clone().contiguous(),
            "v": state.v.detach().clone().contiguous(),
            "h_alt": z(), "u_alt": z(), "v_alt": z(),
            "do_h": state.dh.detach().clone().contiguous(),
            "do_u": state.du.detach().clone().contiguous(),
            "do_v": state.dv.detach().clone().contiguous(),
            "dn_h": z(), "dn_u": z(), "dn_v": z(),
            "fe": z(), "fn": z(), "q": z(), "ke": z(),
        }
        self._fb = fb

    def _swap(self, *names):
        fb = self._fb
        for k in names:
            fb[k], fb[f"{k}_alt"] = fb[f"{k}_alt"], fb[k]

    def _stage_plan(self):
        """Kernel-id triple for the (pre-tendency, tendency, friction)
        stages — static per (dtype, env) so it is resolved once at buffer
        init, not per step.

        float32 runs the vectorized stage kernels (2 columns/thread by
        default — the measured-fastest occupancy point); f64 and the
        MPI4JAX_AMD_SW_NOVEC escape hatch run the scalar ones.  The
        other variants (4-col, nontemporal, two-pass, fully-merged
        scalar) stay selectable for measurement — see
        profiles/README.md for the numbers behind the default.
        """
        import os

        if os.environ.get("MPI4JAX_AMD_SW_MERGED"):
            return None, 8, 7
        if (self.dtype == torch.float32
                and not os.environ.get("MPI4JAX_AMD_SW_NOVEC")):
            if os.environ.get("MPI4JAX_AMD_SW_TWOPASS"):
                return 11, 16, 17
            if os.environ.get("MPI4JAX_AMD_SW_4COL"):
                return None, 18, 17  # 4-col variants (4 waves/SIMD)
            if os.environ.get("MPI4JAX_AMD_SW_NT"):
                return None, 20, 27  # + nontemporal streaming hints
            tile = os.environ.get("MPI4JAX_AMD_SW_TILE", "")
            if tile in ("4", "8", "16"):
                # band-tiled tendency kernel: TJ rows per 256-thread block
                return None, {"4": 21, "8": 22, "16": 23}[tile], 27
            if (self.lateral_viscosity > 0
                    and not os.environ.get("MPI4JAX_AMD_SW_OVERLAP")
                    and os.environ.get("MPI4JAX_AMD_SW_NOFUSE") != "1"):
                # update+friction fused (stage 30 fast kernel + boundary
                # ring kernels) — drops the u'/v' intermediate round
                # trip (4 of 16 field passes); equivalent to the
                # two-kernel path to FMA-contraction rounding
                # (tests/test_gpu_ops.py::test_stage30_*).  s7=None
                # marks the world-1 shape (in-kernel wrap refresh, zero
                # exchange launches); s7=33 the multi-rank shape (a real
                # fe/fn strip exchange between ringA and ringB).
                if (self.grid.nproc_y * self.grid.nproc_x == 1
                        and not self._force_remote_exchange):
                    if os.environ.get("MPI4JAX_AMD_SW_FUSE512") == "1":
                        return None, 31, None  # 512-thread variant
                    return None, 30, None
                return None, 32, 33
            # 2-col merged single pass: 68 VGPRs -> 7 waves/SIMD, measured
            # fastest (stage18v parks ~48% of cycles on memory at 4 waves)
            return None, 19, 27
        return 1, 6, 7

    def _step_fused(self, state, first_step=False):
        from .._backend import rccl

        ext = rccl.ext()
        fb = self._fb
        if fb is None or state.h is not fb["h"]:
            self._init_fused_buffers(state)
            fb = self._fb
        # step-invariant host work is resolved once per buffer (re)init
        const = fb.get("step_const")
        if const is None:
            const = fb["step_const"] = (
                self._fused_flags(),
                float(CORIOLIS_F + float(self.y_local[0]) * CORIOLIS_BETA),
                float(self.dy * CORIOLIS_BETA),
                self._stage_plan(),
            )
        flags, cor_base, cor_dj, (s1, s6, s7) = const
        ab_a, ab_b = ((1.0, 0.0) if first_step
                      else (ADAMS_BASHFORTH_A, ADAMS_BASHFORTH_B))

        def stage(n):
            bufs = [fb["fe"], fb["fn"], fb["q"], fb["ke"], fb["h"],
                    fb["u"], fb["v"], fb["dn_h"], fb["dn_u"], fb["dn_v"],
                    fb["do_h"], fb["do_u"], fb["do_v"], fb["h_alt"],
                    fb["u_alt"], fb["v_alt"]]
            ext.sw_stage(n, bufs, self.dx, self.dy, self.dt,
                         self.lateral_viscosity, cor_base, cor_dj, ab_a,
                         ab_b, flags)

        if s1 is not None:
            stage(s1)     # fe, fn, q, ke (with open-edge halo formulas)
        if s6 in (30, 31):
            # fused update+friction: u'/v' never round-trip through HBM,
            # the mid-step exchange disappears, and the ring kernel
            # writes the end-of-step wrap refresh itself — a complete
            # model step with zero exchange launches
            stage(s6)
            self._swap("h", "u", "v")
            for k in ("h", "u", "v"):
                fb[f"do_{k}"], fb[f"dn_{k}"] = fb[f"dn_{k}"], fb[f"do_{k}"]
            return ModelState(fb["h"], fb["u"], fb["v"], fb["do_h"],
                              fb["do_u"], fb["do_v"])
        if s6 == 32:
            # multi-rank fused step: the friction's only remote need is
            # the u'/v' boundary strip (staged in fe/fn) — exchange
            # those two fields between the ring kernels, then refresh
            # the finals' halos once at the end
            stage(s6)     # fast kernel + ringA (completes the strip)
            self._exchange_fields([fb["fe"], fb["fn"]])
            stage(s7)     # ringB: ring friction from the strip
            self._swap("h", "u", "v")
            self._exchange_fields([fb["h"], fb["u"], fb["v"]])
            for k in ("h", "u", "v"):
                fb[f"do_{k}"], fb[f"dn_{k}"] = fb[f"dn_{k}"], fb[f"do_{k}"]
            return ModelState(fb["h"], fb["u"], fb["v"], fb["do_h"],
                              fb["do_u"], fb["do_v"])
        stage(s6)         # tendencies + time update -> h_alt/u_alt/v_alt
        self._swap("h", "u", "v")
        if self._overlap_plan() is not None and self.lateral_viscosity > 0:
            # halo/compute overlap (MPI4JAX_AMD_SW_OVERLAP=1): run the
            # h/u/v exchange on a second stream while the main stream
            # computes the halo-independent friction pairs (stage 28);
            # the halo-dependent ring (stage 29) joins after.  28+29 use
            # the identical per-pair code as stage 27, so the step is
            # bitwise equal to the serial path.
            s28, s29, comm_stream, ev1, ev2 = self._overlap_plan()
            main = torch.cuda.current_stream()
            ev1.record(main)
            comm_stream.wait_event(ev1)
            with torch.cuda.stream(comm_stream):
                self._exchange_fields([fb["h"], fb["u"], fb["v"]])
            stage(s28)    # deep interior, overlapped with the exchange
            ev2.record(comm_stream)
            main.wait_event(ev2)
            stage(s29)    # boundary ring, needs the fresh halos
            self._swap("u", "v")
            self._exchange_fields([fb["u"], fb["v"]])
        else:
            self._exchange_fields([fb["h"], fb["u"], fb["v"]])
            if self.lateral_viscosity > 0:
                stage(s7)     # friction Laplacian update -> u_alt/v_alt
                self._swap("u", "v")
                self._exchange_fields([fb["u"], fb["v"]])
        # the new tendencies become "old" for the next step
        for k in ("h", "u", "v"):
            fb[f"do_{k}"], fb[f"dn_{k}"] = fb[f"dn_{k}"], fb[f"do_{k}"]
        return ModelState(fb["h"], fb["u"], fb["v"], fb["do_h"], fb["do_u"],
                          fb["do_v"])

    # ------------------------------------------------------------------
    def _overlap_plan(self):
        """(interior_stage, ring_stage, comm_stream) when the overlap path
        is enabled and applicable (float32 vector path, GPU), else None.
        Cached with the fused buffers."""
        import os

        fb = self._fb
        if "overlap_plan" in fb:
            return fb["overlap_plan"]
        plan = None
        if (os.environ.get("MPI4JAX_AMD_SW_OVERLAP") == "1"
                and self.dtype == torch.float32
                and self.device.type == "cuda"
                and self._stage_plan()[2] == 27):
            plan = (28, 29, torch.cuda.Stream(), torch.cuda.Event(),
                    torch.cuda.Event())
        fb["overlap_plan"] = plan
        return plan

    def _exchange_cache(self):
        """Step-invariant halo-exchange state: the resolved schedule (as
        flat int lists for the one-call C++ executor), the staging
        buffers, and the RCCL handle.  Built once per fused-buffer init.
        """
        fb = self._fb
        cache = fb.get("xcache")
        if cache is not None:
            return cache
        ny, nx = self.ny_local, self.nx_local
        for k in ("col_sbuf0", "col_rbuf0", "col_sbuf1", "col_rbuf1"):
            fb[k] = torch.empty(3 * ny, dtype=self.dtype,
                                device=self.device)
        fb["cor_sbuf"] = torch.empty(12, dtype=self.dtype,
                                     device=self.device)
        fb["cor_rbuf"] = torch.empty_like(fb["cor_sbuf"])
        (wrap_sides, col_ops, row_ops, cor_ops,
         cor_mask) = halo_exchange_schedule(self.grid, nx, ny)
        if getattr(self, "_force_remote_exchange", False) and wrap_sides:
            # test hook: express the periodic self-wraps as remote
            # transfers to self so the full RCCL exchange path (pack,
            # grouped p2p, unpack) runs at world 1 — used by the overlap
            # and loopback GPU tests (see tests/test_gpu_overlap.py)
            me = self.comm.rank
            wrap_sides = []
            col_ops = [(0, me, me, 1, nx - 1), (1, me, me, nx - 2, 0)]
        z = -1  # C++ encoding of "no peer"

        def flat(ops):
            return [z if v is None else int(v) for op in ops for v in op]

        has_remote = bool(col_ops or row_ops or cor_ops)
        comm_id = self.comm.rccl_handle() if has_remote else -1
        cache = fb["xcache"] = {
            "sched": (wrap_sides, col_ops, row_ops, cor_ops, cor_mask),
            "flat": (list(wrap_sides), flat(col_ops), flat(row_ops),
                     flat(cor_ops), cor_mask),
            "col_bufs": [fb["col_sbuf0"], fb["col_rbuf0"],
                         fb["col_sbuf1"], fb["col_rbuf1"]],
            "comm_id": comm_id,
        }
        return cache

    def _exchange_fields(self, fields):
        """One-group halo exchange (halo_plan is order-independent).

        Self-wrap columns run as a kernel (no RCCL — keeps the world-1
        path graph-capturable with zero comm init); everything remote —
        packed columns, in-place interior rows, diagonal corners — goes
        into a SINGLE RCCL group enqueue.  The whole exchange is one
        C++ call (``sw_exchange``) so the non-graph multi-rank loop is
        not host-bound; MPI4JAX_AMD_SW_PYEXCHANGE selects the
        wire-identical per-op Python executor instead.
        """
        import os

        from .._backend import rccl

        cache = self._exchange_cache()
        if os.environ.get("MPI4JAX_AMD_SW_PYEXCHANGE"):
            return self._exchange_fields_py(fields, cache)
        fb = self._fb
        wrap_sides, col_flat, row_flat, cor_flat, cor_mask = cache["flat"]
        rccl.ext().sw_exchange(
            fields, wrap_sides, col_flat, row_flat, cor_flat, cor_mask,
            cache["col_bufs"], fb["cor_sbuf"], fb["cor_rbuf"],
            cache["comm_id"],
        )

    def _exchange_fields_py(self, fields, cache):
        """Per-op executor for the halo schedule (reference/debugging —
        identical message set and enqueue order to ``sw_exchange``)."""
        from .._backend import rccl

        ext = rccl.ext()
        nf = len(fields)
        ny, nx = self.ny_local, self.nx_local
        fb = self._fb
        wrap_sides, col_ops, row_ops, cor_ops, cor_mask = cache["sched"]

        for side in wrap_sides:
            ext.halo_wrap(fields, side)
        for k, st, rf, sidx, _ in col_ops:
            if st is not None:
                ext.pack_cols(fb[f"col_sbuf{k}"][:nf * ny], fields, sidx)
        if cor_ops:
            ext.pack_corners(fb["cor_sbuf"][:4 * nf], fields)

        if not (col_ops or row_ops or cor_ops):
            return
        comm_id = cache["comm_id"]
        ext.group_start()
        for k, st, rf, _, _ in col_ops:
            if st is not None:
                ext.send(fb[f"col_sbuf{k}"][:nf * ny], st, comm_id)
            if rf is not None:
                ext.recv(fb[f"col_rbuf{k}"][:nf * ny], rf, comm_id)
        for st, rf, ridx, sidx in row_ops:
            for f in fields:
                if st is not None:
                    ext.send(f[sidx, 1:nx - 1], st, comm_id)
                if rf is not None:
                    ext.recv(f[ridx, 1:nx - 1], rf, comm_id)
        for d, st, rf in cor_ops:
            if st is not None:
                ext.send(fb["cor_sbuf"][d * nf:(d + 1) * nf], st, comm_id)
            if rf is not None:
                ext.recv(fb["cor_rbuf"][d * nf:(d + 1) * nf], rf, comm_id)
        ext.group_end()

        for k, st, rf, _, ridx in col_ops:
            if rf is not None:
                ext.unpack_cols(fields, fb[f"col_rbuf{k}"][:nf * ny], ridx)
        if cor_mask:
            ext.unpack_corners(fields, fb["cor_rbuf"][:4 * nf], cor_mask)

    # ------------------------------------------------------------------
    def make_stepper(self, state, steps_per_call=2, use_graph=None):
        """Return ``(advance, current)``: ``advance()`` runs
        ``steps_per_call`` model steps and returns the current state.

        On the fused GPU path the steps are captured into a single
        hipGraph (halo-exchange RCCL enqueues included — validated by
        tools/probe_rccl_graph.py), so a whole multistep replays with one
        launch from the host.  ``steps_per_call`` must be even (buffer
        parity).  At any world size the capture is adopted only after a
        replay reproduces the eager trajectory bitwise, with all-or-none
        agreement across ranks; otherwise every rank falls back to the
        (wire-identical) eager loop.  ``MPI4JAX_AMD_SW_GRAPH=0`` forces
        the eager loop.
        """
        import os

        env = os.environ.get("MPI4JAX_AMD_SW_GRAPH", "").strip()
        if use_graph is None:
            use_graph = (self.fused and self.device.type == "cuda"
                         and env != "0")
        if (use_graph and self.fused and self.device.type == "cuda"
                and self._fb is not None
                and self._overlap_plan() is not None):
            # hipGraph capture of the two-stream overlapped step segfaults
            # inside the HIP runtime on ROCm 7.0 (not catchable by the
            # validate-then-adopt protocol) — the overlap path targets the
            # non-graph multi-rank loop; capture is disabled with it.
            use_graph = False
        if use_graph and steps_per_call % 2:
            raise ValueError("steps_per_call must be even for graph capture")

        def eager_stepper(s):
            holder = {"s": s}

            def advance():
                for _ in range(steps_per_call):
                    holder["s"] = self.step(holder["s"])
                return holder["s"]

            return advance, s

        if not use_graph:
            return eager_stepper(state)

        # make sure the fused buffers exist & are warm (2 steps, also
        # pre-allocating every temp the exchange path uses and
        # establishing every RCCL p2p connection the capture will record)
        state = self.step(state)
        state = self.step(state)
        torch.cuda.synchronize()

        # Validate-then-adopt protocol (all-or-none across ranks): record
        # the eager trajectory, rewind, capture the same steps into a
        # hipGraph, replay once and require a bitwise match.  Both
        # agreement points are plain allreduces so no rank ever replays a
        # graph while another runs the eager loop with a half-captured
        # schedule; on any failure every rank rewinds and uses the eager
        # loop (wire-identical per step, so even that path stays matched).
        fb = self._fb
        keys = ("h", "u", "v", "do_h", "do_u", "do_v")
        s0 = {k: fb[k].clone() for k in keys}
        s = state
        for _ in range(steps_per_call):
            s = self.step(s)
        torch.cuda.synchronize()
        ref = {k: fb[k].clone() for k in keys}
        for k in keys:
            fb[k].copy_(s0[k])
        torch.cuda.synchronize()

        try:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                s = state
                for _ in range(steps_per_call):
                    s = self.step(s)
            final = s
            captured = True
        except Exception:
            captured = False
            torch.cuda.synchronize()

        if not self._all_ranks_agree(captured):
            for k in keys:
                fb[k].copy_(s0[k])
            torch.cuda.synchronize()
            return eager_stepper(state)

        graph.replay()
        torch.cuda.synchronize()
        matches = all(torch.equal(fb[k], ref[k]) for k in keys)
        # rewind so the net effect of make_stepper is always exactly the
        # two warm-up steps, whichever path was adopted
        for k in keys:
            fb[k].copy_(s0[k])
        torch.cuda.synchronize()
        if not self._all_ranks_agree(matches):
            return eager_stepper(state)

        def advance():
            graph.replay()
            return final

        return advance, state

    def _all_ranks_agree(self, ok):
        """MIN-allreduce of a local predicate over the model's comm."""
        if self.comm.size == 1:
            return bool(ok)
        from ..ops.allreduce import allreduce
        from ..ops.reduce_ops import MIN

        flag = torch.tensor([1 if ok else 0], dtype=torch.int32,
                            device=self.device)
        return bool(allreduce(flag, MIN, comm=self.comm).item() == 1)

    def _step_eager(self, state, first_step=False):
        h, u, v, dh, du, dv = state
        dx, dy = self.dx, self.dy
        eb = self.enforce_boundaries

        hc = self._edge_pad_interior(h)
        hc = eb(hc, "h")

        # mass fluxes on cell faces
        fe = torch.zeros_like(u)
        fn = torch.zeros_like(u)
        fe[_I, _I] = 0.5 * (hc[_I, _I] + hc[_I, _R]) * u[_I, _I]
        fn[_I, _I] = 0.5 * (hc[_I, _I] + hc[_R, _I]) * v[_I, _I]
        fe = eb(fe, "u")
        fn = eb(fn, "v")

        dh_new = torch.zeros_like(dh)
        dh_new[_I, _I] = (
            -(fe[_I, _I] - fe[_I, _L]) / dx - (fn[_I, _I] - fn[_L, _I]) / dy
        )

        # potential vorticity (planetary + relative, / layer thickness)
        q = torch.zeros_like(u)
        q[_I, _I] = self.coriolis[_I, _I] + (
            (v[_I, _R] - v[_I, _I]) / dx - (u[_R, _I] - u[_I, _I]) / dy
        )
        q[_I, _I] *= 1.0 / (
            0.25 * (hc[_I, _I] + hc[_I, _R] + hc[_R, _I] + hc[_R, _R])
        )
        q = eb(q, "h")

        du_new = torch.zeros_like(du)
        dv_new = torch.zeros_like(dv)
        du_new[_I, _I] = (
            -GRAVITY * (h[_I, _R] - h[_I, _I]) / dx
            + 0.5 * (
                q[_I, _I] * 0.5 * (fn[_I, _I] + fn[_I, _R])
                + q[_L, _I] * 0.5 * (fn[_L, _I] + fn[_L, _R])
            )
        )
        dv_new[_I, _I] = (
            -GRAVITY * (h[_R, _I] - h[_I, _I]) / dy
            - 0.5 * (
                q[_I, _I] * 0.5 * (fe[_I, _I] + fe[_R, _I])
                + q[_I, _L] * 0.5 * (fe[_I, _L] + fe[_R, _L])
            )
        )

        # kinetic energy gradient
        ke = torch.zeros_like(u)
        ke[_I, _I] = 0.5 * (
            0.5 * (u[_I, _I] ** 2 + u[_I, _L] ** 2)
            + 0.5 * (v[_I, _I] ** 2 + v[_L, _I] ** 2)
        )
        ke = eb(ke, "h")
        du_new[_I, _I] -= (ke[_I, _R] - ke[_I, _I]) / dx
        dv_new[_I, _I] -= (ke[_R, _I] - ke[_I, _I]) / dy

        # time integration
        h = h.clone()
        u = u.clone()
        v = v.clone()
        dtt = self.dt
        if first_step:
            u[_I, _I] += dtt * du_new[_I, _I]
            v[_I, _I] += dtt * dv_new[_I, _I]
            h[_I, _I] += dtt * dh_new[_I, _I]
        else:
            u[_I, _I] += dtt * (ADAMS_BASHFORTH_A * du_new[_I, _I]
                                + ADAMS_BASHFORTH_B * du[_I, _I])
            v[_I, _I] += dtt * (ADAMS_BASHFORTH_A * dv_new[_I, _I]
                                + ADAMS_BASHFORTH_B * dv[_I, _I])
            h[_I, _I] += dtt * (ADAMS_BASHFORTH_A * dh_new[_I, _I]
                                + ADAMS_BASHFORTH_B * dh[_I, _I])

        h = eb(h, "h")
        u = eb(u, "u")
        v = eb(v, "v")

        # lateral friction
        if self.lateral_viscosity > 0:
            nu = self.lateral_viscosity
            gu = torch.zeros_like(u)
            gv = torch.zeros_like(u)
            gu[_I, _I] = nu * (u[_I, _R] - u[_I, _I]) / dx
            gv[_I, _I] = nu * (u[_R, _I] - u[_I, _I]) / dy
            gu = eb(gu, "u")
            gv = eb(gv, "v")
            u = u.clone()
            u[_I, _I] += dtt * ((gu[_I, _I] - gu[_I, _L]) / dx
                                + (gv[_I, _I] - gv[_L, _I]) / dy)
            gu = torch.zeros_like(v)
            gv = torch.zeros_like(v)
            if self.ref_friction_bug:
                # the reference demo's formula verbatim (mixes v and u:
                # examples/shallow_water.py:386-391) — compat mode only
                gu[_I, _I] = nu * (v[_I, _R] - u[_I, _I]) / dx
                gv[_I, _I] = nu * (v[_R, _I] - u[_I, _I]) / dy
            else:
                gu[_I, _I] = nu * (v[_I, _R] - v[_I, _I]) / dx
                gv[_I, _I] = nu * (v[_R, _I] - v[_I, _I]) / dy
            gu = eb(gu, "u")
            gv = eb(gv, "v")
            v = v.clone()
            v[_I, _I] += dtt * ((gu[_I, _I] - gu[_I, _L]) / dx
                                + (gv[_I, _I] - gv[_L, _I]) / dy)
            # keep u/v halos fresh after the viscosity increment (the
            # reference leaves them one-substep stale; refreshing them is
            # both cleaner physics and what lets the fused GPU path compute
            # derived-field halos locally, bitwise-equal to an exchange)
            u = eb(u, "u")
            v = eb(v, "v")

        return ModelState(h, u, v, dh_new, du_new, dv_new)

    # ------------------------------------------------------------------
    def solve(self, t1_seconds, num_multisteps=100, state=None,
              collect=False):
        """Iterate to t1; returns (final_state, steps_taken, wall_seconds)."""
        if state is None:
            state = self.initial_conditions()
        sol = [state] if collect else None
        state = self.step(state, first_step=True)
        steps = 1
        t = self.dt

        # hipGraph-captured multistep when nothing inspects intermediate
        # states (collect=False); graph replays alias the model buffers,
        # so the collecting path stays on the eager loop
        advance = None
        if (not collect and self.fused and num_multisteps % 2 == 0
                and t + self.dt * num_multisteps < t1_seconds):
            advance, state = self.make_stepper(
                state, steps_per_call=num_multisteps
            )
            steps += 2  # make_stepper warm-up steps
            t += 2 * self.dt

        if self.device.type == "cuda":
            torch.cuda.synchronize()
        start = time.perf_counter()
        while t < t1_seconds:
            if advance is not None:
                state = advance()
            else:
                for _ in range(num_multisteps):
                    state = self.step(state)
            steps += num_multisteps
            t += self.dt * num_multisteps
            if collect:
                sol.append(state)
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        wall = time.perf_counter() - start
        if collect:
            return state, steps, wall, sol
        return state, steps, wall

    def steps_per_model_day(self):
        return DAY_IN_SECONDS / self.dt

    def gather_global(self, field):
        """Gather a distributed field's interior onto rank 0 (debug/plot)."""
        from ..ops.gather import gather

        parts = gather(field[_I, _I].contiguous(), root=0, comm=self.comm)
        if self.comm.rank != 0:
            return None
        npy, npx = self.grid.nproc_y, self.grid.nproc_x
        ny_l, nx_l = self.ny_local - 2, self.nx_local - 2
        out = parts.reshape(npy, npx, ny_l, nx_l)
        return out.permute(0, 2, 1, 3).reshape(npy * ny_l, npx * nx_l)
