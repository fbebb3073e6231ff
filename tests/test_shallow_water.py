"""Shallow-water model tests.

The strongest check: a domain-decomposed 4-rank (2×2) run must reproduce
the single-rank whole-domain solution — this exercises the full ordered
halo-exchange path (sendrecv/send/recv) exactly like the reference demo
(SURVEY.md §3.4).
"""

import torch

import mpi4jax_amd as m
from mpi4jax_amd.models import ShallowWater
from tests._mp import run_multiproc


def test_single_process_runs_and_is_finite():
    sw = ShallowWater(nx=36, ny=18)
    state, steps, wall = sw.solve(t1_seconds=sw.dt * 20, num_multisteps=5)
    assert steps >= 20
    for f in state:
        assert torch.isfinite(f).all()
    # mass conservation-ish: mean height stays near DEPTH
    assert abs(state.h[1:-1, 1:-1].mean().item() - 100.0) < 1.0


def test_steps_per_model_day():
    sw = ShallowWater(nx=36, ny=18)
    assert 4000 < sw.steps_per_model_day() < 4600  # dt ≈ 19.95 s


def _decomposed_matches_single(rank, ws):
    n_steps = 12
    # distributed 2x2 run
    sw = ShallowWater(nx=24, ny=12, comm=m.get_world(), dims=(2, 2))
    state = sw.initial_conditions()
    state = sw.step(state, first_step=True)
    for _ in range(n_steps - 1):
        state = sw.step(state)
    h_full = sw.gather_global(state.h)
    u_full = sw.gather_global(state.u)

    # single-rank whole-domain run on a singleton communicator
    solo_comm = m.get_world().Split(color=rank)
    solo = ShallowWater(nx=24, ny=12, comm=solo_comm, dims=(1, 1))
    sstate = solo.initial_conditions()
    sstate = solo.step(sstate, first_step=True)
    for _ in range(n_steps - 1):
        sstate = solo.step(sstate)

    if rank == 0:
        assert h_full is not None
        ref_h = sstate.h[1:-1, 1:-1]
        ref_u = sstate.u[1:-1, 1:-1]
        assert torch.allclose(h_full, ref_h, atol=1e-5, rtol=1e-5), (
            (h_full - ref_h).abs().max()
        )
        assert torch.allclose(u_full, ref_u, atol=1e-5, rtol=1e-5)


def test_decomposed_matches_single_2x2():
    run_multiproc(_decomposed_matches_single, 4)


def _clockwise_oracle(g, arr):
    """The reference demo's exchange (full edges, clockwise W,N,E,S with
    progressive updates — shallow_water.py:180-208), used as the oracle
    for the order-independent halo_plan implementation."""
    me = g.comm.rank
    out = arr.clone()
    edges = {
        "west": (lambda a: a[:, 1], lambda a, v: a.__setitem__(
            (slice(None), 0), v)),
        "east": (lambda a: a[:, -2], lambda a, v: a.__setitem__(
            (slice(None), -1), v)),
        "south": (lambda a: a[1, :], lambda a, v: a.__setitem__(
            (0, slice(None)), v)),
        "north": (lambda a: a[-2, :], lambda a, v: a.__setitem__(
            (-1, slice(None)), v)),
    }
    for sdir, rdir in (("west", "east"), ("north", "south"),
                       ("east", "west"), ("south", "north")):
        st, rf = g.neighbor(sdir), g.neighbor(rdir)
        if st is None and rf is None:
            continue
        get_s, _ = edges[sdir]
        get_r, set_r = edges[rdir]
        if st == me and rf == me:
            set_r(out, get_s(out))
        elif st is None:
            set_r(out, m.recv(get_r(out), source=rf, comm=g.comm))
        elif rf is None:
            m.send(get_s(out).contiguous(), dest=st, comm=g.comm)
        else:
            set_r(out, m.sendrecv(get_s(out).contiguous(), get_r(out),
                                  source=rf, dest=st, comm=g.comm))
    return out


def _halo_plan_vs_oracle(rank, ws):
    """The order-independent plan (cols + interior rows + diagonal
    corners) must reproduce the reference's clockwise full-edge exchange
    exactly — corners included — for every periodic-x topology."""
    from mpi4jax_amd.parallel.grid import CartesianGrid

    for dims in ((2, 2), (2, 1), (1, 2), (ws, 1), (1, ws)):
        if dims[0] * dims[1] != ws:
            continue
        g = CartesianGrid(m.get_world(), dims=dims, periodic=(False, True))
        torch.manual_seed(100 + rank)
        for ny, nx in ((6, 7), (5, 5)):
            a = torch.randn(ny, nx)
            got = g.halo_exchange(a)
            exp = _clockwise_oracle(g, a)
            assert torch.equal(got, exp), (dims, rank,
                                           (got - exp).abs().max())


def test_halo_plan_vs_clockwise_oracle_4ranks():
    run_multiproc(_halo_plan_vs_oracle, 4)


def test_halo_plan_vs_clockwise_oracle_2ranks():
    run_multiproc(_halo_plan_vs_oracle, 2)


def _two_rank(rank, ws):
    sw = ShallowWater(nx=24, ny=12, comm=m.get_world(), dims=(2, 1))
    state, steps, wall = sw.solve(t1_seconds=sw.dt * 10, num_multisteps=5)
    assert torch.isfinite(state.h).all()


def test_two_rank_column_decomposition():
    run_multiproc(_two_rank, 2)


class _FakeComm:
    def __init__(self, rank, size):
        self.rank, self.size = rank, size

    def rccl_handle(self):
        return 1000 + self.rank  # placeholder; never dereferenced on CPU


def test_halo_schedule_cross_rank_matching():
    """For every topology the driver can run (1..8 ranks), the fused
    exchange schedule's RCCL group must pair up: for each ordered rank
    pair (a, b), a's send sequence to b must equal b's recv sequence from
    a in count and size.  This is the deadlock/matching property of the
    multi-GPU path, checked without a GPU."""
    from mpi4jax_amd.parallel.grid import (CartesianGrid,
                                           halo_exchange_schedule)

    ny, nx = 8, 9
    NF = 3
    topologies = [(1, 1), (2, 1), (1, 2), (2, 2), (2, 4), (4, 2), (1, 8),
                  (8, 1), (2, 3), (3, 2)]
    for dims in topologies:
        size = dims[0] * dims[1]
        sends = {}  # (src, dst) -> [bytes...]
        recvs = {}  # (dst, src) -> [bytes...]
        for rank in range(size):
            g = CartesianGrid.__new__(CartesianGrid)
            g.comm = _FakeComm(rank, size)
            g.nproc_y, g.nproc_x = dims
            g.periodic_y, g.periodic_x = False, True
            g.coords = (rank // dims[1], rank % dims[1])
            wraps, col_ops, row_ops, cor_ops, mask = \
                halo_exchange_schedule(g, nx, ny)
            # emission order must mirror _exchange_fields exactly
            for _, st, rf, _, _ in col_ops:
                if st is not None:
                    assert st != rank  # self handled by wraps only
                    sends.setdefault((rank, st), []).append(NF * ny)
                if rf is not None:
                    recvs.setdefault((rank, rf), []).append(NF * ny)
            for st, rf, _, _ in row_ops:
                for _f in range(NF):
                    if st is not None:
                        sends.setdefault((rank, st), []).append(nx - 2)
                    if rf is not None:
                        recvs.setdefault((rank, rf), []).append(nx - 2)
            for d, st, rf in cor_ops:
                if st is not None:
                    sends.setdefault((rank, st), []).append(NF)
                if rf is not None:
                    recvs.setdefault((rank, rf), []).append(NF)
        all_pairs = set(sends) | {(s, d) for (d, s) in recvs}
        for (src, dst) in all_pairs:
            tx = sends.get((src, dst), [])
            rx = recvs.get((dst, src), [])
            assert tx == rx, (dims, src, dst, tx, rx)


def test_stage_plan_selection(monkeypatch):
    """Env-var kernel-path selection resolved once at buffer init."""
    import torch
    from mpi4jax_amd.models import ShallowWater

    sw = ShallowWater(nx=12, ny=6, device="cpu")
    for var in ("MPI4JAX_AMD_SW_MERGED", "MPI4JAX_AMD_SW_NOVEC",
                "MPI4JAX_AMD_SW_TWOPASS", "MPI4JAX_AMD_SW_4COL",
                "MPI4JAX_AMD_SW_NT", "MPI4JAX_AMD_SW_NOFUSE"):
        monkeypatch.delenv(var, raising=False)
    # single-rank fully-local halos: the fused update+friction kernel
    assert sw._stage_plan() == (None, 30, None)
    monkeypatch.setenv("MPI4JAX_AMD_SW_NOFUSE", "1")
    assert sw._stage_plan() == (None, 19, 27)
    monkeypatch.delenv("MPI4JAX_AMD_SW_NOFUSE")
    # remote halos (forced) select the multi-rank fused shape: fast
    # kernel + ringA (32), real fe/fn strip exchange, ringB (33)
    sw_r = ShallowWater(nx=12, ny=6, device="cpu",
                        _force_remote_exchange=True)
    assert sw_r._stage_plan() == (None, 32, 33)
    monkeypatch.setenv("MPI4JAX_AMD_SW_NOFUSE", "1")
    monkeypatch.setenv("MPI4JAX_AMD_SW_4COL", "1")
    assert sw._stage_plan() == (None, 18, 17)
    monkeypatch.delenv("MPI4JAX_AMD_SW_4COL")
    monkeypatch.setenv("MPI4JAX_AMD_SW_NT", "1")
    assert sw._stage_plan() == (None, 20, 27)
    monkeypatch.delenv("MPI4JAX_AMD_SW_NT")
    monkeypatch.setenv("MPI4JAX_AMD_SW_TWOPASS", "1")
    assert sw._stage_plan() == (11, 16, 17)
    monkeypatch.delenv("MPI4JAX_AMD_SW_TWOPASS")
    monkeypatch.setenv("MPI4JAX_AMD_SW_NOVEC", "1")
    assert sw._stage_plan() == (1, 6, 7)
    monkeypatch.delenv("MPI4JAX_AMD_SW_NOVEC")
    monkeypatch.setenv("MPI4JAX_AMD_SW_MERGED", "1")
    assert sw._stage_plan() == (None, 8, 7)
    # f64 always takes the scalar kernels
    sw64 = ShallowWater(nx=12, ny=6, device="cpu", dtype=torch.float64)
    monkeypatch.delenv("MPI4JAX_AMD_SW_MERGED")
    assert sw64._stage_plan() == (1, 6, 7)


def test_exchange_cache_flat_encoding():
    """The flattened int schedule handed to the native sw_exchange must
    encode exactly the Python schedule with None -> -1, for every
    topology (the C++ side only executes this encoding)."""
    from mpi4jax_amd.models.shallow_water import ShallowWater
    from mpi4jax_amd.parallel.grid import halo_exchange_schedule

    ny, nx = 8, 10
    for dims in [(1, 1), (2, 1), (2, 2), (2, 4), (3, 2), (1, 8)]:
        size = dims[0] * dims[1]
        for rank in range(size):
            sw = ShallowWater.__new__(ShallowWater)
            import torch as _t

            from mpi4jax_amd.parallel.grid import CartesianGrid
            g = CartesianGrid.__new__(CartesianGrid)
            g.comm = _FakeComm(rank, size)
            g.nproc_y, g.nproc_x = dims
            g.periodic_y, g.periodic_x = False, True
            g.coords = (rank // dims[1], rank % dims[1])
            sw.grid = g
            sw.comm = g.comm
            sw.ny_local, sw.nx_local = ny, nx
            sw.device = _t.device("cpu")
            sw.dtype = _t.float32
            sw._fb = {}
            cache = sw._exchange_cache()
            wrap_s, col_s, row_s, cor_s, mask_s = cache["sched"]
            wrap_f, col_f, row_f, cor_f, mask_f = cache["flat"]
            assert wrap_f == list(wrap_s) and mask_f == mask_s

            def flatten(ops):
                return [-1 if v is None else int(v)
                        for op in ops for v in op]

            assert col_f == flatten(col_s)
            assert row_f == flatten(row_s)
            assert cor_f == flatten(cor_s)
            assert len(col_f) % 5 == 0
            assert len(row_f) % 4 == 0
            assert len(cor_f) % 3 == 0
            # no remote peers at world 1 => comm never resolved
            if size == 1:
                assert cache["comm_id"] == -1


def _agree_worker(rank, ws):
    import torch

    from mpi4jax_amd.models import ShallowWater

    sw = ShallowWater(nx=12, ny=6, device="cpu")
    # unanimous yes
    assert sw._all_ranks_agree(True) is True
    # one dissenter -> everyone sees False (all-or-none)
    assert sw._all_ranks_agree(rank != 1) is False
    # unanimous no
    assert sw._all_ranks_agree(False) is False
    assert isinstance(sw._all_ranks_agree(torch.tensor(True)), bool)


def test_graph_adoption_agreement_cross_rank():
    """The hipGraph validate-then-adopt protocol requires all ranks to
    reach the same decision; its agreement reduction must be all-or-none
    across ranks (gloo analog of the RCCL path)."""
    from tests._mp import run_multiproc

    run_multiproc(_agree_worker, 2)


def test_ref_friction_bug_compat_flag():
    """docs/PARITY.md: ref_friction_bug=True reproduces the reference
    demo's v-friction formula (examples/shallow_water.py:386-391, which
    mixes v and u); the default fixes it.  The two trajectories must
    diverge (the term is nonzero) and both stay finite."""
    import pytest
    from mpi4jax_amd.models import ShallowWater

    torch.manual_seed(0)
    fixed = ShallowWater(nx=24, ny=12, device="cpu", dtype=torch.float64)
    compat = ShallowWater(nx=24, ny=12, device="cpu", dtype=torch.float64,
                          ref_friction_bug=True)
    sf = fixed.initial_conditions()
    sc = compat.initial_conditions()
    assert torch.equal(sf.h, sc.h)
    sf = fixed.step(sf, first_step=True)
    sc = compat.step(sc, first_step=True)
    for _ in range(5):
        sf = fixed.step(sf)
        sc = compat.step(sc)
    assert torch.isfinite(sf.v).all() and torch.isfinite(sc.v).all()
    assert not torch.equal(sf.v, sc.v)
    with pytest.raises(ValueError):
        ShallowWater(nx=24, ny=12, ref_friction_bug=True, fused=True)
