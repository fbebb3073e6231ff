"""Multi-rank RCCL probe: N ranks sharing one MI355X (or one rank per GPU).

Exercises every multi-rank native branch in ``csrc/bridge.cpp`` that cannot
run at world_size=1: grouped p2p alltoall/gather/scatter
(bridge.cpp:333-396), the scan ring + combine kernel (:446-468),
send/recv/sendrecv against a real peer (:304-330), ``sw_exchange`` remote
columns/rows/corners (:660-788), and hipGraph capture at world>1.

Launch (2 ranks, one GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --master-port 29571 \
        tools/multirank_gpu_probe.py

Each section prints ``PROBE <name> PASS|FAIL`` from rank 0 so one failure
does not mask the rest; exits nonzero if anything failed.  The final line
``MULTI_RANK_RCCL_OK`` is the marker VERDICT.md round 1 asked for.
"""

import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import mpi4jax_amd as m

RESULTS = []


def section(name):
    def deco(fn):
        RESULTS.append((name, fn))
        return fn

    return deco


@section("allreduce_values")
def _(comm, r, ws):
    for dtype in (torch.float32, torch.bfloat16, torch.float64, torch.int32):
        x = torch.full((1024,), float(r + 1)).to(dtype).cuda()
        y = m.allreduce(x, m.SUM, comm=comm)
        torch.cuda.synchronize()
        expect = sum(range(1, ws + 1))
        assert y[0].item() == expect, (dtype, y[0].item(), expect)
        assert x[0].item() == r + 1  # input immutability
    x = torch.full((16,), float(r + 2), device="cuda")
    assert m.allreduce(x, m.MAX, comm=comm)[0].item() == ws + 1
    assert m.allreduce(x, m.MIN, comm=comm)[0].item() == 2
    import math

    assert m.allreduce(x, m.PROD, comm=comm)[0].item() == math.factorial(ws + 1)


@section("allgather_bcast")
def _(comm, r, ws):
    x = torch.full((3, 4), float(r), device="cuda")
    y = m.allgather(x, comm=comm)
    torch.cuda.synchronize()
    assert y.shape == (ws, 3, 4)
    for k in range(ws):
        assert y[k, 0, 0].item() == k
    b = torch.full((8,), float(r * 10), device="cuda")
    z = m.bcast(b, 1, comm=comm)
    torch.cuda.synchronize()
    assert z[0].item() == 10.0, z[0].item()


@section("alltoall_grouped_p2p")
def _(comm, r, ws):
    # row i of rank r carries value r*ws + i; after alltoall rank r holds
    # column r: entry i == i*ws + r
    x = (torch.arange(ws, device="cuda", dtype=torch.float32) + r * ws)
    x = x[:, None].expand(ws, 5).contiguous()
    y = m.alltoall(x, comm=comm)
    torch.cuda.synchronize()
    for i in range(ws):
        assert y[i, 0].item() == i * ws + r, (i, y[i, 0].item())


@section("gather_scatter_reduce")
def _(comm, r, ws):
    x = torch.full((2, 3), float(r + 1), device="cuda")
    g = m.gather(x, 0, comm=comm)
    torch.cuda.synchronize()
    if r == 0:
        assert g.shape == (ws, 2, 3)
        for k in range(ws):
            assert g[k, 0, 0].item() == k + 1
    src = torch.stack([torch.full((4,), float(10 + k)) for k in range(ws)]).cuda()
    s = m.scatter(src if r == 0 else torch.empty(4, device="cuda"), 0,
                  comm=comm)
    torch.cuda.synchronize()
    assert s[0].item() == 10 + r, s[0].item()
    red = m.reduce(torch.full((6,), float(r + 1), device="cuda"), m.SUM, 0,
                   comm=comm)
    torch.cuda.synchronize()
    if r == 0:
        assert red[0].item() == sum(range(1, ws + 1))


@section("scan_ring_combine_kernel")
def _(comm, r, ws):
    for dtype in (torch.float32, torch.int64, torch.bfloat16):
        x = torch.full((257,), float(r + 1)).to(dtype).cuda()
        y = m.scan(x, m.SUM, comm=comm)
        torch.cuda.synchronize()
        expect = sum(range(1, r + 2))
        assert y[0].item() == expect, (dtype, r, y[0].item(), expect)
    x = torch.full((31,), float(ws - r), device="cuda")
    y = m.scan(x, m.MIN, comm=comm)
    torch.cuda.synchronize()
    assert y[0].item() == min(ws - k for k in range(r + 1))


@section("send_recv_peer")
def _(comm, r, ws):
    from mpi4jax_amd.utils.status import Status

    peer = r ^ 1
    if peer < ws:
        x = torch.full((100,), float(r + 7), device="cuda")
        st = Status()
        if r % 2 == 0:
            m.send(x, peer, comm=comm)
            y = m.recv(torch.empty_like(x), peer, comm=comm, status=st)
        else:
            y = m.recv(torch.empty_like(x), peer, comm=comm, status=st)
            m.send(x, peer, comm=comm)
        torch.cuda.synchronize()
        assert y[0].item() == peer + 7, y[0].item()
        assert st.source == peer


@section("sendrecv_ring")
def _(comm, r, ws):
    dest = (r + 1) % ws
    src = (r - 1) % ws
    x = torch.full((64, 3), float(r), device="cuda")
    y = m.sendrecv(x, x, source=src, dest=dest, comm=comm)
    torch.cuda.synchronize()
    assert y[0, 0].item() == src, y[0, 0].item()


@section("sw_exchange_fused_vs_eager")
def _(comm, r, ws):
    from mpi4jax_amd.models import ShallowWater
    from mpi4jax_amd.parallel.grid import default_dims

    dims = default_dims(ws)
    torch.manual_seed(0)
    kw = dict(nx=128, ny=96, comm=comm, dims=dims, device="cuda",
              dtype=torch.float64)
    fused = ShallowWater(fused=True, **kw)
    eager = ShallowWater(fused=False, **kw)
    sf = fused.initial_conditions()
    se = eager.initial_conditions()
    assert torch.allclose(sf.h, se.h)
    sf = fused.step(sf, first_step=True)
    se = eager.step(se, first_step=True)
    for _ in range(20):
        sf = fused.step(sf)
        se = eager.step(se)
    torch.cuda.synchronize()
    for name in ("h", "u", "v"):
        a, b = getattr(sf, name), getattr(se, name)
        err = (a - b).abs().max().item()
        assert err < 1e-9, (name, err)


@section("graph_capture_world_n")
def _(comm, r, ws):
    from mpi4jax_amd.models import ShallowWater
    from mpi4jax_amd.parallel.grid import default_dims

    dims = default_dims(ws)
    sw = ShallowWater(nx=256, ny=128, comm=comm, dims=dims, device="cuda",
                      dtype=torch.float32)
    state = sw.initial_conditions()
    state = sw.step(state, first_step=True)
    for _ in range(3):
        state = sw.step(state)
    advance, state = sw.make_stepper(state, steps_per_call=4)
    for _ in range(5):
        state = advance()
    torch.cuda.synchronize()
    assert torch.isfinite(state.h).all()


@section("allreduce_busbw_2rank")
def _(comm, r, ws):
    import time

    nbytes = 64 * 1024 * 1024
    x = torch.randn(nbytes // 2, device="cuda").to(torch.bfloat16)
    for _ in range(5):
        m.allreduce(x, m.SUM, comm=comm)
    m.barrier(comm=comm)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        m.allreduce(x, m.SUM, comm=comm)
    m.barrier(comm=comm)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 20
    algbw = nbytes / dt / 1e9
    busbw = algbw * 2 * (ws - 1) / ws
    return {"algbw_GBps": round(algbw, 1), "busbw_GBps": round(busbw, 1),
            "note": "ranks share one GPU unless N matches device count"}


def main():
    m.init(device=int(os.environ.get("PROBE_DEVICE", "0"))
           if torch.cuda.device_count() == 1
           else None)
    comm = m.get_world()
    r, ws = comm.rank, comm.size
    assert ws > 1, "run under torchrun with nproc>1"
    failed = []
    for name, fn in RESULTS:
        dist.barrier()
        try:
            extra = fn(comm, r, ws)
            torch.cuda.synchronize()
            ok = True
        except Exception:
            ok = False
            extra = None
            traceback.print_exc()
        oks = [None] * ws
        dist.all_gather_object(oks, ok)
        if r == 0:
            status = "PASS" if all(oks) else f"FAIL {oks}"
            print(f"PROBE {name} {status} {extra if extra else ''}",
                  flush=True)
        if not all(oks):
            failed.append(name)
    if r == 0:
        if failed:
            print(f"MULTI_RANK_RCCL_FAILED: {failed}", flush=True)
        else:
            print(f"MULTI_RANK_RCCL_OK world_size={ws}", flush=True)
    sys.exit(1 if failed else 0)


if __name__ == "__main__":
    main()
