"""Direct unit tests for the halo-engine kernels.

These kernels otherwise run only inside *remote* exchanges (multi-GPU),
so pin their semantics against plain torch indexing here at world 1.
Layouts must match parallel/grid.py halo_plan:
pack buffers are [field-major per column] for cols and [diag d][field f]
for corners with send cells d0 (1,1), d1 (1,nx-2), d2 (ny-2,1),
d3 (ny-2,nx-2) and recv cells d0 (ny-1,nx-1), d1 (ny-1,0), d2 (0,nx-1),
d3 (0,0).
"""

import pytest
import torch

import mpi4jax_amd as m  # noqa: F401  (init side effects)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    import mpi4jax_amd._rccl_C as e

    return e


def fields(ny=7, nx=9, nf=3, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(ny, nx, generator=g).to(dtype).cuda()
            for _ in range(nf)]


@pytest.mark.parametrize("nf", [1, 2, 3])
@pytest.mark.parametrize("col", [0, 1, 7, 8])
def test_pack_cols(ext, nf, col):
    fs = fields(nf=nf)
    ny = fs[0].shape[0]
    buf = torch.empty(nf * ny, device="cuda")
    ext.pack_cols(buf, fs, col)
    torch.cuda.synchronize()
    expect = torch.cat([f[:, col] for f in fs])
    assert torch.equal(buf, expect)


@pytest.mark.parametrize("col", [0, 8])
def test_unpack_cols(ext, col):
    fs = fields()
    ny = fs[0].shape[0]
    originals = [f.clone() for f in fs]
    data = torch.randn(3 * ny, device="cuda")
    ext.unpack_cols(fs, data, col)
    torch.cuda.synchronize()
    for i, (f, orig) in enumerate(zip(fs, originals)):
        assert torch.equal(f[:, col], data[i * ny:(i + 1) * ny])
        mask = torch.ones_like(f, dtype=torch.bool)
        mask[:, col] = False
        assert torch.equal(f[mask], orig[mask])  # nothing else touched


def test_pack_corners_layout(ext):
    fs = fields(nf=3)
    ny, nx = fs[0].shape
    buf = torch.empty(12, device="cuda")
    ext.pack_corners(buf, fs)
    torch.cuda.synchronize()
    send_cells = [(1, 1), (1, nx - 2), (ny - 2, 1), (ny - 2, nx - 2)]
    for d, (j, i) in enumerate(send_cells):
        for f_idx, f in enumerate(fs):
            assert buf[d * 3 + f_idx].item() == f[j, i].item(), (d, f_idx)


@pytest.mark.parametrize("mask", [0b1111, 0b0101, 0b0010, 0])
def test_unpack_corners_mask(ext, mask):
    fs = fields(nf=2)
    ny, nx = fs[0].shape
    originals = [f.clone() for f in fs]
    data = torch.arange(8.0, device="cuda")
    ext.unpack_corners(fs, data, mask)
    torch.cuda.synchronize()
    recv_cells = [(ny - 1, nx - 1), (ny - 1, 0), (0, nx - 1), (0, 0)]
    for d, (j, i) in enumerate(recv_cells):
        for f_idx, f in enumerate(fs):
            if mask & (1 << d):
                assert f[j, i].item() == data[d * 2 + f_idx].item()
            else:
                assert f[j, i].item() == originals[f_idx][j, i].item()


def test_halo_wrap_sides(ext):
    for side in (0, 1, 2):
        fs = fields(nf=3)
        originals = [f.clone() for f in fs]
        ext.halo_wrap(fs, side)
        torch.cuda.synchronize()
        for f, orig in zip(fs, originals):
            if side in (0, 2):
                assert torch.equal(f[:, -1], orig[:, 1])
            else:
                assert torch.equal(f[:, -1], orig[:, -1])
            if side in (1, 2):
                assert torch.equal(f[:, 0], orig[:, -2])
            else:
                assert torch.equal(f[:, 0], orig[:, 0])
            assert torch.equal(f[:, 1:-1], orig[:, 1:-1])


def test_pack_cols_f64(ext):
    fs = fields(dtype=torch.float64, nf=2)
    ny = fs[0].shape[0]
    buf = torch.empty(2 * ny, dtype=torch.float64, device="cuda")
    ext.pack_cols(buf, fs, 3)
    torch.cuda.synchronize()
    assert torch.equal(buf, torch.cat([f[:, 3] for f in fs]))


# ---------------------------------------------------------------------------
# sw_exchange: the one-call C++ executor of the halo schedule.  At world 1
# only the wrap/pack plumbing can run (no remote peers), so pin that against
# plain indexing and against the wire-identical Python executor.

def _bufs(ny, nf=3, dtype=torch.float32):
    col = [torch.zeros(3 * ny, device="cuda", dtype=dtype)
           for _ in range(4)]
    cs = torch.zeros(12, device="cuda", dtype=dtype)
    cr = torch.zeros(12, device="cuda", dtype=dtype)
    return col, cs, cr


@pytest.mark.parametrize("wrap", [[2], [0], [1], [0, 1], []])
@pytest.mark.parametrize("nf", [2, 3])
def test_sw_exchange_wraps(ext, wrap, nf):
    fs = fields(nf=nf)
    expect = [f.clone() for f in fs]
    for side in wrap:
        for f in expect:
            if side in (0, 2):
                f[:, -1] = f[:, 1]
            if side in (1, 2):
                f[:, 0] = f[:, -2]
    ny = fs[0].shape[0]
    col, cs, cr = _bufs(ny, nf)
    ext.sw_exchange(fs, wrap, [], [], [], 0, col, cs, cr, -1)
    torch.cuda.synchronize()
    for f, e in zip(fs, expect):
        assert torch.equal(f, e)


def test_sw_exchange_packs_without_peers(ext):
    # a schedule whose only entries are sends-with-no-peer / recv-with-no-
    # peer must still run the pack kernels and never touch the comm
    # (comm_id -1 would raise if looked up)
    fs = fields()
    ny, nx = fs[0].shape
    col, cs, cr = _bufs(ny)
    # col op k=0: send to nobody, recv from nobody -> pack skipped, no comm
    ext.sw_exchange(fs, [], [0, -1, -1, 1, nx - 1], [], [], 0, col, cs, cr,
                    -1)
    torch.cuda.synchronize()
    assert torch.equal(col[0], torch.zeros_like(col[0]))


def test_sw_exchange_bad_schedule_raises(ext):
    fs = fields()
    ny = fs[0].shape[0]
    col, cs, cr = _bufs(ny)
    with pytest.raises(RuntimeError, match="schedule"):
        ext.sw_exchange(fs, [], [0, -1, -1], [], [], 0, col, cs, cr, -1)


def test_fused_model_python_executor_matches(monkeypatch):
    """The per-op Python executor and the one-call C++ executor must
    produce bitwise-identical model trajectories (same kernels, same
    message set; at world 1 that covers the wrap path end to end)."""
    from mpi4jax_amd.models import ShallowWater

    def run(pyexchange):
        if pyexchange:
            monkeypatch.setenv("MPI4JAX_AMD_SW_PYEXCHANGE", "1")
        else:
            monkeypatch.delenv("MPI4JAX_AMD_SW_PYEXCHANGE", raising=False)
        sw = ShallowWater(nx=72, ny=36, device="cuda")
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(5):
            s = sw.step(s)
        torch.cuda.synchronize()
        return s

    a = run(False)
    b = run(True)
    for f in ("h", "u", "v"):
        assert torch.equal(getattr(a, f), getattr(b, f)), f


# ---------------------------------------------------------------------------
# merged staging kernels (pack_halo / unpack_halo): one launch must equal
# the single-purpose kernels run in sequence.

@pytest.mark.parametrize("wrap", [-1, 0, 1, 2])
@pytest.mark.parametrize("nf", [1, 3])
def test_pack_halo_matches_separate(ext, wrap, nf):
    fs = fields(nf=nf)
    ref = [f.clone() for f in fs]
    ny, nx = fs[0].shape
    cb0 = torch.zeros(nf * ny, device="cuda")
    cb1 = torch.zeros(nf * ny, device="cuda")
    cor = torch.zeros(4 * nf, device="cuda")
    ext.pack_halo(fs, wrap, cb0, 1, cb1, nx - 2, cor)

    e_cb0 = torch.zeros_like(cb0)
    e_cb1 = torch.zeros_like(cb1)
    e_cor = torch.zeros_like(cor)
    if wrap >= 0:
        ext.halo_wrap(ref, wrap)
    ext.pack_cols(e_cb0, ref, 1)
    ext.pack_cols(e_cb1, ref, nx - 2)
    ext.pack_corners(e_cor, ref)
    torch.cuda.synchronize()
    assert torch.equal(cb0, e_cb0)
    assert torch.equal(cb1, e_cb1)
    assert torch.equal(cor, e_cor)
    for f, r in zip(fs, ref):
        assert torch.equal(f, r)  # wrap applied identically


def test_pack_halo_skips_null_segments(ext):
    fs = fields()
    before = [f.clone() for f in fs]
    ny = fs[0].shape[0]
    cb1 = torch.full((3 * ny,), -7.0, device="cuda")
    ext.pack_halo(fs, -1, None, 0, cb1, 1, None)
    torch.cuda.synchronize()
    for f, b in zip(fs, before):
        assert torch.equal(f, b)
    assert not torch.equal(cb1, torch.full_like(cb1, -7.0))


@pytest.mark.parametrize("cor_mask", [0, 0b1111, 0b0101])
def test_unpack_halo_matches_separate(ext, cor_mask):
    nf = 3
    fs = fields(nf=nf)
    ref = [f.clone() for f in fs]
    ny, nx = fs[0].shape
    g = torch.Generator().manual_seed(7)
    cb0 = torch.randn(nf * ny, generator=g).cuda()
    cb1 = torch.randn(nf * ny, generator=g).cuda()
    cor = torch.randn(4 * nf, generator=g).cuda()
    ext.unpack_halo(fs, cb0, nx - 1, cb1, 0, cor if cor_mask else None,
                    cor_mask)

    # reference order: columns land first, corners overwrite
    ext.unpack_cols(ref, cb0, nx - 1)
    ext.unpack_cols(ref, cb1, 0)
    if cor_mask:
        ext.unpack_corners(ref, cor, cor_mask)
    torch.cuda.synchronize()
    for f, r in zip(fs, ref):
        assert torch.equal(f, r)


def test_v2_stage_variant_matches(monkeypatch):
    """The 2-column default kernels (stages 19/27) must reproduce the
    4-column variants (same operation order; float rounding only)."""
    from mpi4jax_amd.models import ShallowWater

    def run(wide):
        if wide:
            monkeypatch.setenv("MPI4JAX_AMD_SW_4COL", "1")
        else:
            monkeypatch.delenv("MPI4JAX_AMD_SW_4COL", raising=False)
        sw = ShallowWater(nx=130, ny=66, device="cuda")
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(8):
            s = sw.step(s)
        torch.cuda.synchronize()
        return s

    a = run(False)
    b = run(True)
    for f in ("h", "u", "v"):
        x, y = getattr(a, f), getattr(b, f)
        assert torch.allclose(x, y, atol=1e-5, rtol=1e-6), (
            f, (x - y).abs().max().item())


def test_v2_stage_variant_walls(monkeypatch):
    """2-col kernels with closed east boundary (wall masks + scalar
    edge columns)."""
    from mpi4jax_amd.models import ShallowWater

    def run(wide):
        if wide:
            monkeypatch.setenv("MPI4JAX_AMD_SW_4COL", "1")
        else:
            monkeypatch.delenv("MPI4JAX_AMD_SW_4COL", raising=False)
        sw = ShallowWater(nx=97, ny=49, device="cuda", periodic_x=False)
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(6):
            s = sw.step(s)
        torch.cuda.synchronize()
        return s

    a = run(False)
    b = run(True)
    for f in ("h", "u", "v"):
        x, y = getattr(a, f), getattr(b, f)
        assert torch.allclose(x, y, atol=1e-5, rtol=1e-6), (
            f, (x - y).abs().max().item())


def test_nt_stage_variant_matches(monkeypatch):
    """The nontemporal-hint kernel (stage 20) is a pure cache-policy
    change: results must be bitwise equal to the default stage 19."""
    from mpi4jax_amd.models import ShallowWater

    def run(nt):
        # compare against stage 19 (the NT kernel is its cache-policy
        # twin) — not the fused stage 30 world-1 default
        monkeypatch.setenv("MPI4JAX_AMD_SW_NOFUSE", "1")
        if nt:
            monkeypatch.setenv("MPI4JAX_AMD_SW_NT", "1")
        else:
            monkeypatch.delenv("MPI4JAX_AMD_SW_NT", raising=False)
        sw = ShallowWater(nx=130, ny=66, device="cuda")
        s = sw.initial_conditions()
        s = sw.step(s, first_step=True)
        for _ in range(6):
            s = sw.step(s)
        torch.cuda.synchronize()
        return s

    a = run(False)
    b = run(True)
    for f in ("h", "u", "v"):
        assert torch.equal(getattr(a, f), getattr(b, f)), f
