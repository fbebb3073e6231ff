"""torch.compile integration: collectives trace with fullgraph=True."""

import pytest
import torch

import mpi4jax_amd as m
from mpi4jax_amd import jit_ops


def test_allreduce_compiles_fullgraph():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, "sum") * 2

    x = torch.arange(4.0)
    y = f(x)
    assert torch.equal(y, x * 2)


def test_allreduce_compiled_grad():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, "sum").sum()

    x = torch.randn(5, requires_grad=True)
    f(x).backward()
    assert torch.equal(x.grad, torch.ones(5))


def test_allgather_compiled_shape():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allgather(x)

    y = f(torch.zeros(3, 2))
    assert y.shape == (1, 3, 2)


def test_sendrecv_compiled_self_and_grad():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.sendrecv(x, x.detach(), source=0, dest=0).sum()

    x = torch.randn(4, requires_grad=True)
    f(x).backward()
    assert torch.equal(x.grad, torch.ones(4))


def test_scan_alltoall_bcast_compiled():
    @torch.compile(fullgraph=True)
    def f(x):
        a = jit_ops.scan(x, "sum")
        b = jit_ops.alltoall(x[None])[0]
        c = jit_ops.bcast(x, 0)
        d = jit_ops.reduce_scatter(x[None], "sum")
        return a + b + c + d

    x = torch.arange(6.0)
    assert torch.equal(f(x), 4 * x)


def test_nonsum_compiled_grad_raises():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, "max").sum()

    x = torch.randn(3, requires_grad=True)
    with pytest.raises(RuntimeError, match="only differentiable"):
        f(x).backward()


@pytest.mark.gpu
def test_compiled_allreduce_gpu():
    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, "sum") + 1

    x = torch.arange(8.0, device="cuda")
    y = f(x)
    torch.cuda.synchronize()
    assert torch.equal(y, x + 1)


def test_reduce_gather_scatter_compiled():
    @torch.compile(fullgraph=True)
    def f(x):
        r = jit_ops.reduce(x, "sum", 0)
        g = jit_ops.gather(x, 0)[0]
        s = jit_ops.scatter(x[None], 0)
        return r + g + s

    x = torch.arange(5.0)
    assert torch.equal(f(x), 3 * x)


@pytest.mark.gpu
def test_compiled_ops_gpu_roundtrip():
    @torch.compile(fullgraph=True)
    def f(x):
        a = jit_ops.scan(x, "sum")
        b = jit_ops.sendrecv(x, x.detach(), source=0, dest=0)
        c = jit_ops.reduce_scatter(x[None], "sum")
        return a + b + c

    x = torch.arange(6.0, device="cuda")
    y = f(x)
    torch.cuda.synchronize()
    assert torch.equal(y, 3 * x)


def test_comm_key_recompiles_per_communicator(monkeypatch):
    """Communicators cross compiled graphs as int keys computed OUTSIDE
    the graph; keys are runtime data, so one compiled function serves any
    communicator with the correct registry entry at runtime."""
    import mpi4jax_amd as m
    from mpi4jax_amd.ops import jit_ops as jo

    seen = []
    orig = jo._comm
    monkeypatch.setattr(jo, "_comm", lambda k: (seen.append(k), orig(k))[1])

    ca = m.get_world().Clone()
    cb = m.get_world().Clone()
    ka, kb = jo.comm_key(ca), jo.comm_key(cb)

    @torch.compile(fullgraph=True)
    def f(x, k):
        return jo.allreduce(x, "sum", comm=k)

    x = torch.ones(4)
    f(x, ka)
    f(x, kb)
    assert len(seen) >= 2
    assert seen[-1] == kb and seen[0] == ka, (seen, ka, kb)


def test_eager_ops_inside_compile_graph_break_ok():
    """The eager API inside torch.compile (without fullgraph) must still
    be correct — dynamo graph-breaks around it and falls back to eager."""
    @torch.compile  # no fullgraph: breaks are allowed
    def f(x):
        return m.allreduce(x * 2, m.SUM) + 1

    x = torch.arange(4.0)
    assert torch.equal(f(x), 2 * x + 1)


def test_compiled_bitwise_allreduce():
    """Bitwise ops trace through the custom-op layer (static op arg)."""
    import mpi4jax_amd as m
    from mpi4jax_amd import jit_ops

    @torch.compile(fullgraph=True)
    def f(x):
        return jit_ops.allreduce(x, m.BXOR)

    x = torch.tensor([0b101, 0b011], dtype=torch.int32)
    assert torch.equal(f(x), x)  # world 1: identity


@pytest.mark.gpu
def test_compiled_shallow_water_eager_step_gpu():
    """torch.compile over the whole eager shallow-water step (graph breaks
    allowed at the comm ops) must reproduce the uncompiled trajectory."""
    from mpi4jax_amd.models import ShallowWater

    torch.manual_seed(0)
    ref = ShallowWater(nx=64, ny=32, device="cuda", dtype=torch.float64,
                       fused=False)
    cmp_ = ShallowWater(nx=64, ny=32, device="cuda", dtype=torch.float64,
                        fused=False)
    compiled_step = torch.compile(cmp_.step)
    sr = ref.initial_conditions()
    sc = cmp_.initial_conditions()
    sr = ref.step(sr, first_step=True)
    sc = compiled_step(sc, first_step=True)
    for _ in range(3):
        sr = ref.step(sr)
        sc = compiled_step(sc)
    torch.cuda.synchronize()
    for name in ("h", "u", "v"):
        a, b = getattr(sr, name), getattr(sc, name)
        assert torch.allclose(a, b, atol=1e-12), name


@pytest.mark.gpu
def test_vmap_allreduce_gpu():
    x = torch.randn(4, 100, device="cuda")
    y = torch.func.vmap(lambda v: jit_ops.allreduce(v, "sum"))(x)
    torch.cuda.synchronize()
    assert torch.equal(y, x)
    w = torch.randn(4, 100, device="cuda", requires_grad=True)
    torch.func.vmap(lambda v: jit_ops.allreduce(v, "sum"))(w).sum().backward()
    torch.cuda.synchronize()
    assert torch.equal(w.grad, torch.ones_like(w))
