"""Property fuzz: the GPU (RCCL) backend must agree with the CPU (gloo)
backend on every op, shape and dtype at world size 1 (both are exact
degenerate collectives, so results must match bitwise)."""

import pytest
import torch

import mpi4jax_amd as m

pytestmark = pytest.mark.gpu

DTYPES = [torch.float32, torch.float64, torch.float16, torch.bfloat16,
          torch.int8, torch.uint8, torch.int16, torch.int32, torch.int64,
          torch.bool, torch.complex64]

SHAPES = [(), (1,), (7,), (3, 5), (2, 3, 4), (1, 1, 1), (128, 65)]


def _rand(shape, dtype):
    g = torch.Generator().manual_seed(hash((shape, str(dtype))) % (2**31))
    if dtype == torch.bool:
        return torch.rand(shape, generator=g) > 0.5
    if dtype.is_complex:
        return torch.randn(shape, generator=g, dtype=dtype)
    if dtype.is_floating_point:
        return torch.randn(shape, generator=g).to(dtype)
    return torch.randint(-4, 9, shape, generator=g).to(dtype)


@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("shape", SHAPES)
def test_fuzz_allreduce_allgather(shape, dtype):
    x = _rand(shape, dtype)
    xg = x.cuda()
    ops = [m.SUM] if (dtype.is_complex or dtype == torch.bool) else \
        [m.SUM, m.MIN, m.MAX]
    for op in ops:
        a = m.allreduce(x, op)
        b = m.allreduce(xg, op)
        torch.cuda.synchronize()
        assert torch.equal(a, b.cpu()), (shape, dtype, op)
    a = m.allgather(x)
    b = m.allgather(xg)
    torch.cuda.synchronize()
    assert torch.equal(a, b.cpu())


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16,
                                   torch.int64, torch.bool])
@pytest.mark.parametrize("shape", [(4,), (3, 5), (2, 3, 4)])
def test_fuzz_movement_ops(shape, dtype):
    x = _rand(shape, dtype)
    xg = x.cuda()
    pairs = [
        (m.bcast(x, 0), m.bcast(xg, 0)),
        (m.gather(x, 0), m.gather(xg, 0)),
        (m.scatter(x[None], 0), m.scatter(xg[None], 0)),
        (m.reduce(x, m.MAX, 0) if dtype != torch.bool else m.reduce(x, m.MAX, 0),
         m.reduce(xg, m.MAX, 0)),
        (m.scan(x, m.MAX) if dtype != torch.bool else m.scan(x, m.MAX),
         m.scan(xg, m.MAX)),
        (m.sendrecv(x, x, source=0, dest=0),
         m.sendrecv(xg, xg, source=0, dest=0)),
        (m.alltoall(x[None]), m.alltoall(xg[None])),
    ]
    torch.cuda.synchronize()
    for a, b in pairs:
        assert torch.equal(torch.as_tensor(a), torch.as_tensor(b).cpu()), (
            shape, dtype)


def test_fuzz_stage30_random_sizes(monkeypatch):
    """Randomized domain sweep of the fused stage-30 step against the
    two-kernel pipeline (fixed seed for reproducibility): odd/tiny/flag
    shapes beyond the curated sizes in test_gpu_ops.  Sizes reach the
    sub-8-row / sub-12-col ring-enumeration branches and both wall
    configs."""
    import random

    from mpi4jax_amd.models import ShallowWater

    rng = random.Random(20260914)
    sizes = [(rng.randrange(6, 180), rng.randrange(3, 90))
             for _ in range(8)]
    for nx, ny in sizes:
        for periodic_x in (True, False):
            results = {}
            for nofuse in ("1", ""):
                if nofuse:
                    monkeypatch.setenv("MPI4JAX_AMD_SW_NOFUSE", nofuse)
                else:
                    monkeypatch.delenv("MPI4JAX_AMD_SW_NOFUSE",
                                       raising=False)
                sw = ShallowWater(nx=nx, ny=ny, device="cuda", fused=True,
                                  periodic_x=periodic_x,
                                  comm=m.get_world().Clone())
                st = sw.step(sw.initial_conditions(), first_step=True)
                for _ in range(4):
                    st = sw.step(st)
                torch.cuda.synchronize()
                results[nofuse] = {k: getattr(st, k).clone()
                                   for k in ("h", "u", "v")}
            for k in ("h", "u", "v"):
                a, b = results["1"][k], results[""][k]
                assert torch.allclose(a, b, atol=1e-5, rtol=1e-5), (
                    nx, ny, periodic_x, k, (a - b).abs().max().item())
