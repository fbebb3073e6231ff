"""Host-enqueue overhead of the non-graph fused step loop.

The driver's multi-GPU scaling bench runs the fused step WITHOUT hipGraph
capture (graphs are world-1 only for now), so per-step host time bounds
strong scaling: once per-rank compute shrinks below the Python/pybind
enqueue cost, adding GPUs stops helping.  This measures that bound at
world 1 for the full benchmark domain and for a proxy of the N=8 local
domain, for both halo-exchange executors (one-call C++ ``sw_exchange``
vs the per-op Python loop).

Run on a GPU box:  python benchmarks/bench_step_overhead.py
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

from mpi4jax_amd.models import ShallowWater  # noqa: E402


def measure(nx, ny, steps=200):
    sw = ShallowWater(nx=nx, ny=ny, device="cuda", dtype=torch.float32)
    s = sw.initial_conditions()
    s = sw.step(s, first_step=True)
    for _ in range(20):
        s = sw.step(s)
    torch.cuda.synchronize()

    # wall time per step, host in the loop every step (no graph)
    t0 = time.perf_counter()
    for _ in range(steps):
        s = sw.step(s)
    torch.cuda.synchronize()
    wall_ms = (time.perf_counter() - t0) / steps * 1e3

    # pure enqueue time: submit a short burst without waiting on the GPU
    # (50 steps ≈ 500 launches stays well under stream queue limits)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        s = sw.step(s)
    host_ms = (time.perf_counter() - t0) / 50 * 1e3
    torch.cuda.synchronize()
    return round(wall_ms, 4), round(host_ms, 4)


def main():
    out = {}
    for tag, env in (("cpp_exchange", None),
                     ("py_exchange", "MPI4JAX_AMD_SW_PYEXCHANGE")):
        if env:
            os.environ[env] = "1"
        for nx, ny, name in ((3600, 1800, "full_3600x1800"),
                             (900, 900, "n8_local_900x900")):
            wall, host = measure(nx, ny)
            out[f"{tag}/{name}"] = {"wall_ms_per_step": wall,
                                    "host_enqueue_ms_per_step": host}
        if env:
            del os.environ[env]
    print(json.dumps(out, indent=2))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/step_overhead.json", "w") as f:
        json.dump(out, f, indent=2)


if __name__ == "__main__":
    main()
