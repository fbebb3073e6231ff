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


def _phase_schedule_worker(rank, ws):
    """Validate the fused halo phase plan against the eager exchange.

    The fused GPU path (ShallowWater._exchange_fields) follows
    halo_phase_schedule(); executing the same plan with CPU ops must
    reproduce CartesianGrid.halo_exchange exactly, corners included.
    """
    from mpi4jax_amd.parallel.grid import CartesianGrid
    from mpi4jax_amd.models.shallow_water import halo_phase_schedule

    for dims, periodic_x in (((2, 2), True), ((2, 2), False),
                             ((2, 1), True), ((1, 2), True)):
        if dims[0] * dims[1] != ws:
            continue
        g = CartesianGrid(m.get_world(), dims=dims,
                          periodic=(False, periodic_x))
        torch.manual_seed(100 + rank)
        ny, nx = 6, 7
        f0 = torch.randn(ny, nx)
        f1 = torch.randn(ny, nx)
        expect = [g.halo_exchange(f0), g.halo_exchange(f1)]

        fields = [f0.clone(), f1.clone()]
        me = g.comm.rank
        for cols, sdir, rdir, ridx, sidx in halo_phase_schedule(nx, ny):
            st, rf = g.neighbor(sdir), g.neighbor(rdir)
            if st is None and rf is None:
                continue
            if st == me and rf == me:
                for f in fields:
                    if cols:
                        f[:, ridx] = f[:, sidx]
                    else:
                        f[ridx, :] = f[sidx, :]
                continue
            if cols:
                sbuf = torch.cat([f[:, sidx] for f in fields])
                tmpl = torch.empty(len(fields) * ny)
                if st is not None and rf is not None:
                    rbuf = m.sendrecv(sbuf, tmpl, source=rf, dest=st,
                                      comm=g.comm)
                elif st is not None:
                    m.send(sbuf, st, comm=g.comm)
                    rbuf = None
                else:
                    rbuf = m.recv(tmpl, rf, comm=g.comm)
                if rf is not None:
                    for i, f in enumerate(fields):
                        f[:, ridx] = rbuf[i * ny:(i + 1) * ny]
            else:
                for f in fields:
                    if st is not None and rf is not None:
                        got = m.sendrecv(f[sidx, :], f[ridx, :], source=rf,
                                         dest=st, comm=g.comm)
                        f[ridx, :] = got
                    elif st is not None:
                        m.send(f[sidx, :].contiguous(), st, comm=g.comm)
                    else:
                        f[ridx, :] = m.recv(f[ridx, :], rf, comm=g.comm)
        for got, exp in zip(fields, expect):
            assert torch.equal(got, exp), (dims, periodic_x, rank,
                                           (got - exp).abs().max())


def test_fused_halo_phase_schedule_2x2():
    run_multiproc(_phase_schedule_worker, 4)


def test_fused_halo_phase_schedule_2ranks():
    run_multiproc(_phase_schedule_worker, 2)


def _two_rank(rank, ws):
    sw = ShallowWater(nx=24, ny=12, comm=m.get_world(), dims=(2, 1))
    state, steps, wall = sw.solve(t1_seconds=sw.dt * 10, num_multisteps=5)
    assert torch.isfinite(state.h).all()


def test_two_rank_column_decomposition():
    run_multiproc(_two_rank, 2)
