"""Band-tile experiment: correctness + speed of stage 21/22/23 vs 19.

The tiled kernels run the identical per-cell code (stage19_cells), so
their trajectories must be BITWISE equal to the default; the question is
only latency hiding.  Prints ms/step per variant at the benchmark domain.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import mpi4jax_amd as m  # noqa: E402
from mpi4jax_amd.models import ShallowWater  # noqa: E402


def traj(tile, steps=10):
    # compare against the two-kernel stage-19 path (stage 30 fuses the
    # friction and is compared separately in tests/test_gpu_ops.py)
    os.environ["MPI4JAX_AMD_SW_NOFUSE"] = "1"
    if tile:
        os.environ["MPI4JAX_AMD_SW_TILE"] = tile
    else:
        os.environ.pop("MPI4JAX_AMD_SW_TILE", None)
    torch.manual_seed(0)
    sw = ShallowWater(nx=512, ny=256, device="cuda", dtype=torch.float32)
    s = sw.initial_conditions()
    s = sw.step(s, first_step=True)
    for _ in range(steps):
        s = sw.step(s)
    torch.cuda.synchronize()
    return {k: getattr(s, k).clone() for k in ("h", "u", "v")}


def bench(tile, nx=3600, ny=1800, steps=300, warmup=50):
    os.environ["MPI4JAX_AMD_SW_NOFUSE"] = "1"
    if tile:
        os.environ["MPI4JAX_AMD_SW_TILE"] = tile
    else:
        os.environ.pop("MPI4JAX_AMD_SW_TILE", None)
    sw = ShallowWater(nx=nx, ny=ny, device="cuda", dtype=torch.float32)
    s = sw.initial_conditions()
    s = sw.step(s, first_step=True)
    for _ in range(warmup):
        s = sw.step(s)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        s = sw.step(s)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    assert torch.isfinite(s.h).all()
    return dt * 1e3


def main():
    m.init()
    base = traj("")
    for tile in ("4", "8", "16"):
        t = traj(tile)
        ok = all(torch.equal(base[k], t[k]) for k in base)
        print(f"tile {tile}: bitwise equal to default = {ok}", flush=True)
        assert ok, f"tile {tile} diverged"
    for tile in ("", "4", "8", "16"):
        ms = bench(tile)
        print(f"variant {'default(19)' if not tile else 'TJ=' + tile}: "
              f"{ms:.4f} ms/step", flush=True)


if __name__ == "__main__":
    main()
